"""SLURM launcher with requeue-on-preemption checkpointing.

Parity with the reference's run/submit.py:128-207 (which is non-runnable —
it imports a module that does not exist). Uses submitit when
available; falls back to emitting an sbatch script that wraps torchrun (one
process per GPU over RCCL).
"""

from __future__ import annotations

import argparse
import logging
import os
import sys
import uuid
from pathlib import Path

logger = logging.getLogger("dinov3")


def get_args_parser():
    from ..train.train import get_args_parser as trainer_parser

    parser = argparse.ArgumentParser("DINOv3 MI355X SLURM launcher", parents=[trainer_parser(add_help=False)])
    parser.add_argument("--nodes", type=int, default=1)
    parser.add_argument("--ngpus", type=int, default=8, help="GPUs per node (MI355X: 8)")
    parser.add_argument("--timeout", type=int, default=2800, help="job timeout (minutes)")
    parser.add_argument("--partition", type=str, default="")
    parser.add_argument("--qos", type=str, default="")
    parser.add_argument("--comment", type=str, default="")
    parser.add_argument("--exclude", type=str, default="")
    return parser


def get_shared_folder() -> Path:
    for candidate in (os.environ.get("DINOV3_SHARED_DIR"), "/checkpoint", "/tmp"):
        if candidate and Path(candidate).is_dir():
            p = Path(candidate) / "dinov3_amd_experiments"
            p.mkdir(exist_ok=True, parents=True)
            return p
    raise RuntimeError("no shared folder available")


class Trainer:
    """submitit callable with checkpoint() requeue support."""

    def __init__(self, args):
        self.args = args

    def __call__(self):
        self._setup_env()
        from ..train import train as train_module

        train_module.main(self._train_argv())

    def _train_argv(self):
        argv = []
        if self.args.config_file:
            argv += ["--config-file", self.args.config_file]
        if self.args.output_dir:
            argv += ["--output-dir", self.args.output_dir]
        argv += self.args.opts or []
        return argv

    def _setup_env(self):
        import submitit

        env = submitit.JobEnvironment()
        os.environ["MASTER_ADDR"] = env.hostnames[0]
        os.environ.setdefault("MASTER_PORT", "29500")
        os.environ["RANK"] = str(env.global_rank)
        os.environ["LOCAL_RANK"] = str(env.local_rank)
        os.environ["WORLD_SIZE"] = str(env.num_tasks)
        os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
        logger.info("slurm env: rank %s world %s", env.global_rank, env.num_tasks)

    def checkpoint(self):
        """Called by submitit on preemption: requeue resuming from the latest
        checkpoint (the trainer resumes automatically unless --no-resume)."""
        import submitit

        self.args.no_resume = False
        return submitit.helpers.DelayedSubmission(Trainer(self.args))


def _sbatch_fallback(args) -> str:
    """Emit an sbatch script wrapping torchrun when submitit is unavailable."""
    out = Path(args.output_dir or get_shared_folder() / uuid.uuid4().hex[:8])
    out.mkdir(parents=True, exist_ok=True)
    opts = " ".join(args.opts or [])
    script = f"""#!/bin/bash
#SBATCH --job-name=dinov3_amd
#SBATCH --nodes={args.nodes}
#SBATCH --ntasks-per-node=1
#SBATCH --gpus-per-node={args.ngpus}
#SBATCH --cpus-per-task=96
#SBATCH --time={args.timeout}
#SBATCH --output={out}/slurm-%j.out
#SBATCH --signal=USR2@120
#SBATCH --requeue
{f'#SBATCH --partition={args.partition}' if args.partition else ''}
export HSA_ENABLE_IPC_MODE_LEGACY=0
export MASTER_ADDR=$(scontrol show hostnames $SLURM_JOB_NODELIST | head -n1)
export MASTER_PORT=29500
srun python -m torch.distributed.run \\
  --nnodes={args.nodes} --nproc-per-node={args.ngpus} \\
  --rdzv-backend=c10d --rdzv-endpoint=$MASTER_ADDR:$MASTER_PORT \\
  -m dinov3_amd.train.train --config-file {args.config_file} \\
  --output-dir {out} {opts}
"""
    path = out / "launch.sbatch"
    path.write_text(script)
    return str(path)


def main(argv=None):
    args = get_args_parser().parse_args(argv)
    if not args.output_dir:
        args.output_dir = str(get_shared_folder() / uuid.uuid4().hex[:8])
    try:
        import submitit
    except ImportError:
        path = _sbatch_fallback(args)
        print(f"submitit not installed; wrote sbatch script: {path}")
        print(f"submit with: sbatch {path}")
        return path

    executor = submitit.AutoExecutor(folder=args.output_dir, slurm_max_num_timeout=30)
    kwargs = {}
    if args.partition:
        kwargs["slurm_partition"] = args.partition
    if args.qos:
        kwargs["slurm_qos"] = args.qos
    if args.comment:
        kwargs["slurm_comment"] = args.comment
    if args.exclude:
        kwargs["slurm_exclude"] = args.exclude
    executor.update_parameters(
        mem_gb=0,
        gpus_per_node=args.ngpus,
        tasks_per_node=args.ngpus,
        cpus_per_task=12,
        nodes=args.nodes,
        timeout_min=args.timeout,
        name="dinov3_amd",
        **kwargs,
    )
    job = executor.submit(Trainer(args))
    print(f"submitted job {job.job_id} -> {args.output_dir}")
    return job


if __name__ == "__main__":
    main(sys.argv[1:])
