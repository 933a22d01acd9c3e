from .utils import cat_keep_shapes, count_parameters, fix_random_seeds, uncat_with_shapes

__all__ = ["cat_keep_shapes", "uncat_with_shapes", "fix_random_seeds", "count_parameters"]
