"""SLURM launcher: sbatch fallback script generation + Trainer requeue hook
(reference dinov3_jax/run/submit.py, which imports a missing module and
cannot run; this one must)."""

from dinov3_amd.run.submit import Trainer, get_args_parser, main


def test_sbatch_fallback_script(tmp_path):
    out = str(tmp_path / "job")
    # submitit is not installed in this image -> the fallback path runs
    path = main([
        "--nodes", "2", "--ngpus", "8",
        "--config-file", "dinov3_amd/configs/train/vitl_im1k_lin834.yaml",
        "--output-dir", out,
    ])
    text = open(path).read()
    assert "--nnodes=2" in text and "--nproc-per-node=8" in text
    assert "HSA_ENABLE_IPC_MODE_LEGACY=0" in text
    assert "vitl_im1k_lin834.yaml" in text
    assert "#SBATCH --requeue" in text


def test_trainer_checkpoint_requeues(tmp_path):
    """The submitit checkpoint() hook resubmits with resume enabled."""
    args = get_args_parser().parse_args([
        "--config-file", "dinov3_amd/configs/train/vits_smoke.yaml",
        "--output-dir", str(tmp_path),
    ])
    t = Trainer(args)
    assert hasattr(t, "checkpoint")
