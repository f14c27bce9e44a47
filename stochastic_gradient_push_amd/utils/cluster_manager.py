"""SLURM preemption-aware checkpoint manager.

Parity: reference experiment_utils/cluster_manager.py:24-141.
Mechanism: SIGTERM is logged and ignored; SIGUSR1 (SLURM's pre-preemption
warning) sets a flag tensor which is all-reduced at every checkpoint so
*any* rank's signal triggers a coordinated checkpoint-then-requeue
(`scontrol requeue`) across the job.  Checkpoint filename layout
(`checkpoint_r{rank}_n{world}.pth.tar`, `model_best_*`, optional
`ep{epoch}_` prefix) is part of the reproduced API contract.

(The reference called ``sys.exit`` without importing ``sys`` — a latent
crash on the requeue path; fixed here.)
"""

import os
import shutil
import signal
import sys

import torch
import torch.distributed as dist

from .helpers import make_logger


class ClusterManager:
    """Tracks SLURM signals and owns checkpoint save/requeue.

    If ``world_size > 1`` this assumes ``dist.init_process_group`` was
    called before construction.
    """

    MASTER_RANK = 0
    CHECKPOINT_DIR = None

    @staticmethod
    def set_checkpoint_dir(checkpoint_dir):
        ClusterManager.CHECKPOINT_DIR = checkpoint_dir

    def __init__(self, rank, world_size, state, model_tag="", callback=None,
                 all_workers=False):
        """
        :param rank: this agent's rank
        :param world_size: number of agents
        :param state: dict encoding training state (saved verbatim)
        :param model_tag: tag prefixed to checkpoint file names
        :param callback: optional function to run when SIGUSR1 arrives
        :param all_workers: save every agent's model vs only rank 0's
        """
        assert ClusterManager.CHECKPOINT_DIR is not None
        self.rank = rank
        self.world_size = world_size
        self.state = state
        self.all_workers = all_workers
        self.main_pid = os.getpid()
        self.signal_tensor = torch.zeros(1)
        if torch.cuda.is_available():
            self.signal_tensor = self.signal_tensor.cuda()
        self.logger = make_logger(rank)
        self.callback = callback

        model_rank = rank if all_workers else ClusterManager.MASTER_RANK
        self.model_tag = model_tag
        self.checkpoint_fname = (
            f"checkpoint_r{model_rank}_n{world_size}.pth.tar"
        )
        self.model_best_fname = (
            f"model_best_r{model_rank}_n{world_size}.pth.tar"
        )
        self.checkpoint_fpath = (
            ClusterManager.CHECKPOINT_DIR + self.model_tag
            + self.checkpoint_fname
        )
        self.model_best_fpath = (
            ClusterManager.CHECKPOINT_DIR + self.model_tag
            + self.model_best_fname
        )

        self.install_signal_handlers()

        if self.world_size > 1:
            assert dist.is_initialized()
            self.process_group = dist.new_group(list(range(world_size)))

    def save_checkpoint(self, epoch_id=None, requeue_on_signal=True):
        """Save state (+best copy); if any rank saw SIGUSR1, requeue the
        SLURM job and exit (reference cluster_manager.py:86-118)."""
        if requeue_on_signal and self.world_size > 1:
            dist.all_reduce(self.signal_tensor, group=self.process_group)

        self.logger.info("Saving checkpoint")
        if self.all_workers or self.rank == ClusterManager.MASTER_RANK:
            if epoch_id is None:
                checkpoint_fpath = self.checkpoint_fpath
            else:
                checkpoint_fpath = (
                    ClusterManager.CHECKPOINT_DIR + "ep" + str(epoch_id)
                    + "_" + self.model_tag + self.checkpoint_fname
                )
            torch.save(self.state, checkpoint_fpath)
            if self.state.get("is_best"):
                shutil.copyfile(checkpoint_fpath, self.model_best_fpath)
                self.state["is_best"] = False

        if requeue_on_signal and self.signal_tensor[0] > 0:
            self.logger.info("At least 1 process received SIGUSR1; requeueing")
            if self.rank == 0 and os.getpid() == self.main_pid:
                command = f'scontrol requeue {os.environ["SLURM_JOB_ID"]}'
                self.logger.info("Relaunching: " + command)
                if os.system(command):
                    raise RuntimeError("scontrol requeue failed")
                self.logger.info("New job submitted to the queue")
            self.logger.info("Terminating")
            sys.exit(0)

    def install_signal_handlers(self):
        self.logger.info("Signal handlers installed")
        signal.signal(signal.SIGUSR1, self.SIGUSR1Handler)
        signal.signal(signal.SIGTERM, self.SIGTERMHandler)
        self.signal_handlers_installed = True

    def SIGTERMHandler(self, signum, frame):
        """Ignore SIGTERM; SLURM sends SIGUSR1 ahead of preemption and
        that is the one we act on."""
        self.logger.info("Received SIGTERM")

    def SIGUSR1Handler(self, signum, frame):
        self.logger.info("Received SIGUSR1")
        if self.callback is not None:
            self.callback()
        self.signal_tensor[0] = 1
