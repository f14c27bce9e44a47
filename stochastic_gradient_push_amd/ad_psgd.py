"""BilatGossipDataParallel — asynchronous decentralized SGD (AD-PSGD).

Behavioral parity with reference gossip/ad_psgd.py:36-418, rebuilt
MI355X-first:

* Two-process split exactly as in the reference: the **trainer process**
  computes forward/backward and never touches the network; the **gossip
  process** owns the ``torch.distributed`` world (RCCL rank), applies the
  gradients with its own optimizer, and asynchronously averages
  parameters with one peer at a time via
  :class:`~stochastic_gradient_push_amd.gossiper.BilatPushPull`
  (reference ad_psgd.py:120-133, 252-366).
* Shared state is **two flat tensors** (params, grads) in shared memory —
  not ~161 per-tensor handles (reference ad_psgd.py:99-111) — so every
  hand-off is a single contiguous copy and the gossip-side optimizer is
  one fused SGD-momentum kernel over the whole model
  (:func:`~stochastic_gradient_push_amd.ops.sgd_step_`) instead of a
  torch.optim.SGD loop (reference ad_psgd.py:261-266, 340-341).
* The bilateral merge ``p = (p + p_peer) / 2`` (reference
  ad_psgd.py:357-361) is the fused ``average_`` kernel.

Control protocol (mp.Event quartet + mp.Lock + mp.Value for the learning
rate) matches reference ad_psgd.py:113-119, 220-249.
"""

import os
from typing import Optional

import torch
import torch.distributed as dist
import torch.multiprocessing as mp
from torch.autograd import Variable
from torch.nn.modules import Module

from . import ops
from .gossiper import BilatPushPull
from .mixing import UniformMixing
from .ops.flat import FlatParams
from .utils.helpers import make_logger
from .utils.metering import Meter


class BilatGossipDataParallel(Module):
    """Asynchronous bilateral-gossip model wrapper (API parity: reference
    ad_psgd.py:39-48)."""

    def __init__(
        self,
        module: Module,
        device_ids=None,
        master_addr: Optional[str] = None,
        master_port: Optional[str] = None,
        backend: Optional[str] = None,
        world_size: Optional[int] = None,
        rank: Optional[int] = None,
        graph_class=None,
        mixing_class=UniformMixing,
        num_peers: int = 1,
        comm_device: Optional[torch.device] = None,
        lr: float = 0.1,
        momentum: float = 0.9,
        weight_decay: float = 1e-4,
        nesterov: bool = True,
        verbose: bool = False,
        network_interface_type=None,
        tcp_interface_name=None,
    ):
        super().__init__()
        self.module = module
        self.logger = make_logger(rank, verbose)

        if comm_device is None:
            comm_device = torch.device("cpu")
        if comm_device.type == "cuda" and comm_device.index is None:
            comm_device = torch.device("cuda", torch.cuda.current_device())
        self.__cpu_comm = comm_device.type == "cpu"
        if not self.__cpu_comm:
            # device-resident comm: shared CUDA tensors travel to the
            # gossip process as dmabuf-IPC views of the SAME device
            # memory (torch.multiprocessing CUDA sharing); p2p on device
            # buffers needs RCCL — gloo only does host p2p
            if (backend or "nccl") != "nccl":
                raise ValueError(
                    "comm_device=cuda requires backend='nccl' (RCCL): "
                    "gloo cannot p2p device tensors"
                )
            backend = "nccl"

        self.dist_config = {
            "verbose": verbose,
            "graph_class": graph_class,
            "master_addr": master_addr,
            "master_port": master_port,
            "backend": backend,
            "world_size": world_size,
            "rank": rank,
            "mixing_class": mixing_class,
            "lr": lr,
            "momentum": momentum,
            "nesterov": nesterov,
            "weight_decay": weight_decay,
            "comm_device": comm_device,
            "network_interface_type": network_interface_type,
            "num_peers": num_peers,
        }
        self.num_updates = 0
        self.gossip_enable = True

        # flat views over trainable params/grads in the trainer process
        self.flatp = FlatParams(module, flatten_grads=True)

        # shared flat buffers handed to the gossip process.  CPU: true
        # shared memory.  CUDA: the mp.Queue transfer maps the SAME
        # device allocation into the gossip process via dmabuf IPC
        # (reference ad_psgd.py:72-74,105 only supported pinned CPU —
        # this is the SURVEY C10 MI355X-native mode).
        shared_params = self.flatp.flat.detach().to(comm_device).clone()
        shared_grads = torch.zeros_like(shared_params)
        if self.__cpu_comm:
            shared_params.share_memory_()
            shared_grads.share_memory_()
        self.gossip_params_flat = shared_params
        self.gossip_grads_flat = shared_grads

        ctx = mp.get_context("spawn")
        self.gossip_queue = ctx.Queue()
        self.gossip_lock = ctx.Lock()
        self.gossip_enable_flag = ctx.Event()
        self.train_write_flag = ctx.Event()   # trainer wrote new grads
        self.gossip_read_flag = ctx.Event()   # gossip proc consumed grads
        self.gossip_update_flag = ctx.Event() # lr update request
        self._lr = ctx.Value("f", lr, lock=False)
        self.gossip_enable_flag.set()

        self.gossip_thread = ctx.Process(
            target=BilatGossipDataParallel._gossip_target,
            args=(
                self.dist_config,
                self.gossip_enable_flag,
                self.train_write_flag,
                self.gossip_read_flag,
                self.gossip_update_flag,
                self._lr,
                self.gossip_lock,
                self.gossip_queue,
                tcp_interface_name,
            ),
            daemon=True,
            name="Gossip-Process",
        )
        self.gossip_thread.start()
        self.gossip_queue.put((self.gossip_params_flat, self.gossip_grads_flat))

        self.__register_hooks()

    # -- public API ---------------------------------------------------------

    def update_lr(self, lr: float) -> None:
        """Signal the gossip process to change its optimizer lr (reference
        ad_psgd.py:142-147)."""
        if self._lr.value == lr:
            return
        with self.gossip_lock:
            self._lr.value = lr
        self.gossip_update_flag.set()

    def forward(self, *inputs, **kwargs):
        return self.module(*inputs, **kwargs)

    def train(self, mode: bool = True):
        super().train(mode)
        return self

    def eval(self):
        super().eval()
        self._pull_model()
        return self

    def enable_gossip(self):
        self.gossip_enable = True
        self.gossip_enable_flag.set()

    def disable_gossip(self):
        self.gossip_enable = False
        self.gossip_enable_flag.clear()

    def block(self):
        """No-op, as in the reference (ad_psgd.py:212-215 returns before
        its barrier; the trainer process is not in the dist world)."""
        return

    def sync_comms(self):
        self._pull_model()

    def state_dict(self):
        return {"state_dict": super().state_dict()}

    def load_state_dict(self, load_dict):
        super().load_state_dict(load_dict["state_dict"])
        with self.gossip_lock:
            self.gossip_params_flat.copy_(self.flatp.flat)

    # -- trainer<->gossip hand-off -----------------------------------------

    def _pull_model(self):
        """Copy the gossip process's current params into the module
        (reference ad_psgd.py:220-229) — one flat copy.

        CUDA-comm: the mp.Lock orders HOST threads, not GPU streams in
        two processes, so the read must complete before the lock is
        released (the gossip process syncs its writes the same way)."""
        with self.gossip_lock:
            self.flatp.flat.copy_(self.gossip_params_flat)
            if not self.__cpu_comm:
                torch.cuda.current_stream().synchronize()
        return True

    def _transfer_grads(self):
        """Hand freshly computed grads to the gossip process (reference
        ad_psgd.py:232-249)."""
        self.gossip_read_flag.wait()
        self.gossip_grads_flat.copy_(self.flatp.flat_grad)
        if not self.__cpu_comm:
            # the flag is host-side: the device copy must land before
            # the gossip process is told the grads are ready
            torch.cuda.current_stream().synchronize()
        self.gossip_read_flag.clear()
        self.train_write_flag.set()
        return True

    def communicator_warmup(self):
        """Parity stub (reference ad_psgd.py:414-418 barriers inside the
        dist world; the trainer process has no dist context)."""
        return

    # -- gossip process ------------------------------------------------------

    @staticmethod
    def _gossip_target(
        dist_config, gossip_enable_flag, train_write_flag, gossip_read_flag,
        gossip_update_flag, gossip_lr, gossip_lock, gossip_queue,
        tcp_interface_name,
    ):
        """Comm-process main loop (reference ad_psgd.py:252-366): owns the
        RCCL/Gloo rank, applies grads with a fused SGD, and bilaterally
        averages with one peer per iteration."""
        with torch.no_grad():
            if dist_config["comm_device"].type == "cuda":
                # pin the gossip process to the trainer's GPU before the
                # IPC tensors are materialized or RCCL initializes
                torch.cuda.set_device(dist_config["comm_device"])
            gossip_params, gossip_grads = gossip_queue.get()
            momentum_buf = torch.zeros_like(gossip_params)
            first_step = True

            backend = dist_config["backend"] or "gloo"
            if dist_config["network_interface_type"] == "ethernet":
                if backend == "nccl":
                    if tcp_interface_name is not None:
                        os.environ["NCCL_SOCKET_IFNAME"] = tcp_interface_name
                    os.environ["NCCL_IB_DISABLE"] = "1"
                elif backend == "gloo" and tcp_interface_name is not None:
                    os.environ["GLOO_SOCKET_IFNAME"] = tcp_interface_name

            os.environ["MASTER_ADDR"] = dist_config["master_addr"]
            os.environ["MASTER_PORT"] = str(dist_config["master_port"])
            dist.init_process_group(
                backend=backend,
                world_size=dist_config["world_size"],
                rank=dist_config["rank"],
            )
            logger = make_logger(dist.get_rank(), dist_config["verbose"])

            graph_class = dist_config["graph_class"]
            mixing_class = dist_config["mixing_class"]
            # eager communicator creation barrier (reference
            # ad_psgd.py:295-299)
            dist.barrier()
            graph = graph_class(
                dist_config["rank"], dist_config["world_size"],
                peers_per_itr=dist_config["num_peers"],
            )
            mixing = (
                mixing_class(graph, dist_config["comm_device"])
                if mixing_class else None
            )

            gossiper = BilatPushPull(
                gossip_params, graph=graph, mixing=mixing, logger=logger,
            )
            model_meter = Meter(ptag="Model", stateful=True, csv_format=False)
            gossip_meter = Meter(ptag="Gossip", stateful=True, csv_format=False)
            gossip_read_flag.set()
            lr = dist_config["lr"]
            import time as _time

            while True:
                gossip_enable_flag.wait()

                if gossip_update_flag.is_set():
                    with gossip_lock:
                        lr = gossip_lr.value
                    logger.debug(f"updated lr to {lr}")
                    gossip_update_flag.clear()

                cuda_comm = dist_config["comm_device"].type == "cuda"
                if train_write_flag.is_set():
                    bt = _time.time()
                    with gossip_lock:
                        ops.sgd_step_(
                            gossip_params, gossip_grads, momentum_buf,
                            lr=lr,
                            momentum=dist_config["momentum"],
                            weight_decay=dist_config["weight_decay"],
                            nesterov=dist_config["nesterov"],
                            first_step=first_step,
                        )
                        first_step = False
                        if cuda_comm:
                            torch.cuda.current_stream().synchronize()
                    train_write_flag.clear()
                    gossip_read_flag.set()
                    model_meter.update(_time.time() - bt)
                    logger.debug(model_meter)

                try:
                    bt = _time.time()
                    with gossip_lock:
                        out_msg = gossip_params.clone()
                        if cuda_comm:
                            torch.cuda.current_stream().synchronize()
                    in_msg, completed = gossiper.mix(out_msg)
                    if not isinstance(completed, bool) or completed:
                        with gossip_lock:
                            ops.average_(
                                gossip_params, in_msg.to(gossip_params.device)
                            )
                            if cuda_comm:
                                torch.cuda.current_stream().synchronize()
                    gossip_meter.update(_time.time() - bt)
                    logger.debug(gossip_meter)
                except RuntimeError as e:
                    logger.warning(f"received runtime error {e}")
                    gossiper.clean_msg_buffers_()

    # -- hooks ----------------------------------------------------------------

    def __register_hooks(self):
        """Backward hook: after backward completes, push grads to the
        gossip process and pull the freshest model (reference
        ad_psgd.py:372-412)."""
        queue_hook = self.__make_backward_hook()

        def attach(module, inputs, output):
            out = output
            if isinstance(out, (tuple, list)):
                out = next(
                    (t for t in out if torch.is_tensor(t) and t.requires_grad),
                    None,
                )
            if torch.is_tensor(out) and out.requires_grad:
                out.register_hook(queue_hook)
            return output

        self.register_forward_hook(attach)

    def __make_backward_hook(self):
        def hook(*unused):
            # keep grads wired to the flat buffer
            if not self.flatp.grads_wired():
                self.flatp.rewire_grads()
            if self.gossip_enable:
                self._transfer_grads()
                self._pull_model()

        def queue_hook(*unused):
            Variable._execution_engine.queue_callback(hook)

        return queue_hook
