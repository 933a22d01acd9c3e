from .ssl_meta_arch import SSLMetaArch

__all__ = ["SSLMetaArch"]
