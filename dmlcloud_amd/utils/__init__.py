from . import argparse, git, logging, project, seed, slurm, table, tcp, thirdparty, wandb  # noqa: F401
