#!/usr/bin/env python3
"""SGP / OSGP / D-PSGD / AllReduce-SGD trainer CLI.

MI355X-native re-implementation of the reference trainer
(reference gossip_sgd.py:54-712): same flag surface, CSV log schema
(header at reference gossip_sgd.py:264-274), LR warmup+decay schedule
(508-536), peers-per-iteration schedule (497-505), checkpoint/resume with
mid-epoch sampler fast-forward (356-364), KLDiv one-hot loss (192-198),
and SLURM/MPI rank discovery (599-605) — extended with torchrun env
discovery, a synthetic-data mode (this environment has no dataset
downloads), model/norm selection and the fused flat-buffer SGD.
"""

import argparse
import copy
import os
import socket
import time

import torch
import torch.distributed as dist
import torch.nn as nn

from stochastic_gradient_push_amd import (
    DynamicBipartiteExponentialGraph as DBEGraph,
    DynamicBipartiteLinearGraph as DBLGraph,
    DynamicDirectedExponentialGraph as DDEGraph,
    DynamicDirectedLinearGraph as DDLGraph,
    GossipDataParallel,
    NPeerDynamicDirectedExponentialGraph as NPDDEGraph,
    RingGraph,
    UniformMixing,
)
from stochastic_gradient_push_amd.models import build_resnet
from stochastic_gradient_push_amd.ops.fused_sgd import FusedSGD
from stochastic_gradient_push_amd.utils import Meter, make_logger
from stochastic_gradient_push_amd.utils.cluster_manager import ClusterManager
from stochastic_gradient_push_amd.utils.nic import pin_comm_env

GRAPH_TOPOLOGIES = {
    0: DDEGraph,    # Dynamic Directed Exponential
    1: DBEGraph,    # Dynamic Bipartite Exponential
    2: DDLGraph,    # Dynamic Directed Linear
    3: DBLGraph,    # Dynamic Bipartite Linear
    4: RingGraph,   # Ring
    5: NPDDEGraph,  # N-Peer Dynamic Directed Exponential (default)
    -1: None,
}

MIXING_STRATEGIES = {
    0: UniformMixing,
    -1: None,
}


def str2bool(v):
    return str(v) == "True"


def build_parser():
    p = argparse.ArgumentParser(description="MI355X gossip SGD trainer")
    p.add_argument("--all_reduce", default="False", type=str)
    p.add_argument("--batch_size", default=32, type=int,
                   help="per-agent batch size")
    p.add_argument("--lr", default=0.1, type=float,
                   help="reference lr (for a 256-sample global batch)")
    p.add_argument("--num_dataloader_workers", default=4, type=int)
    p.add_argument("--num_epochs", default=90, type=int)
    p.add_argument("--num_iterations_per_training_epoch", default=-1,
                   type=int, help="truncate the train loop (testing only)")
    p.add_argument("--momentum", default=0.9, type=float)
    p.add_argument("--weight_decay", default=1e-4, type=float)
    p.add_argument("--nesterov", default="False", type=str)
    p.add_argument("--push_sum", default="True", type=str)
    p.add_argument("--graph_type", default=5, type=int,
                   choices=list(GRAPH_TOPOLOGIES))
    p.add_argument("--mixing_strategy", default=0, type=int,
                   choices=list(MIXING_STRATEGIES))
    p.add_argument("--schedule", nargs="+", default=None, type=float,
                   help="lr decay schedule: epoch factor pairs")
    p.add_argument("--peers_per_itr_schedule", nargs="+", type=int,
                   help="epoch num_peers pairs (must include epoch 0)")
    p.add_argument("--overlap", default="False", type=str)
    p.add_argument("--synch_freq", default=0, type=int)
    p.add_argument("--warmup", default="False", type=str)
    p.add_argument("--seed", default=47, type=int)
    p.add_argument("--resume", default="False", type=str)
    p.add_argument("--backend", default="nccl",
                   choices=["nccl", "gloo", "mpi"])
    p.add_argument("--tag", default="", type=str)
    p.add_argument("--print_freq", default=10, type=int)
    p.add_argument("--verbose", default="False", type=str)
    p.add_argument("--train_fast", default="False", type=str)
    p.add_argument("--checkpoint_all", default="True", type=str)
    p.add_argument("--overwrite_checkpoints", default="True", type=str)
    p.add_argument("--master_port", default="40100", type=str)
    p.add_argument("--checkpoint_dir", type=str, default="./checkpoints/")
    p.add_argument("--network_interface_type", default="infiniband",
                   choices=["infiniband", "ethernet", "auto"])
    p.add_argument("--num_itr_ignore", type=int, default=10)
    p.add_argument("--dataset_dir", type=str, default=None)
    p.add_argument("--no_cuda_streams", action="store_true")
    p.add_argument("--hip_graph", default="True", type=str,
                   help="hipGraph-capture the local train step (cuda, "
                        "fused_sgd, lazy-mixing algorithms)")
    # MI355X-native extensions
    p.add_argument("--dataset", default="synthetic",
                   choices=["synthetic", "imagefolder"])
    p.add_argument("--synthetic_size", default=2048, type=int,
                   help="samples per agent epoch in synthetic mode")
    p.add_argument("--model", default="resnet50", type=str)
    p.add_argument("--norm", default="fused", type=str,
                   choices=["fused", "native", "miopen"])
    p.add_argument("--conv_impl", default="miopen", type=str,
                   choices=["auto", "miopen", "gemm", "mfma"],
                   help="conv backend (auto: hand-written MFMA kernels on "
                        "the measured winning shapes, MIOpen elsewhere)")
    p.add_argument("--master_weights", default="auto", type=str,
                   choices=["auto", "on", "off"],
                   help="bf16 working weights over an fp32 master "
                        "(auto: on for cuda + fused_sgd)")
    p.add_argument("--steal_grads", default="auto", type=str,
                   choices=["auto", "on", "off"],
                   help="steal-mode grads + fused multi-gather instead of "
                        "per-param accumulate adds (auto: on for cuda + "
                        "fused_sgd)")
    p.add_argument("--num_classes", default=1000, type=int)
    p.add_argument("--fused_sgd", default="True", type=str,
                   help="use the one-kernel flat-buffer SGD")
    p.add_argument("--device", default=None, choices=[None, "cuda", "cpu"])
    p.add_argument("--image_size", default=224, type=int)
    return p


def discover_rank_world(args):
    """SLURM / OpenMPI / torchrun env discovery (reference
    gossip_sgd.py:599-605 + torchrun)."""
    env = os.environ
    if args.backend == "mpi" and "OMPI_COMM_WORLD_RANK" in env:
        return int(env["OMPI_COMM_WORLD_RANK"]), int(env["OMPI_UNIVERSE_SIZE"])
    if "SLURM_PROCID" in env and "SLURM_NTASKS" in env:
        return int(env["SLURM_PROCID"]), int(env["SLURM_NTASKS"])
    if "RANK" in env and "WORLD_SIZE" in env:
        return int(env["RANK"]), int(env["WORLD_SIZE"])
    return 0, 1


def pairs_to_dict(values, default):
    sched = {}
    values = list(values) if values else list(default)
    for epoch, val in zip(values[0::2], values[1::2]):
        sched[epoch] = val
    return sched


def parse_args(argv=None):
    args = build_parser().parse_args(argv)
    ClusterManager.set_checkpoint_dir(args.checkpoint_dir)
    os.makedirs(args.checkpoint_dir, exist_ok=True)

    for flag in ("resume", "verbose", "train_fast", "nesterov",
                 "checkpoint_all", "warmup", "overlap", "push_sum",
                 "all_reduce", "overwrite_checkpoints", "fused_sgd",
                 "hip_graph"):
        setattr(args, flag, str2bool(getattr(args, flag)))

    args.rank, args.world_size = discover_rank_world(args)
    args.master_addr = os.environ.get(
        "MASTER_ADDR", os.environ.get("HOSTNAME", "127.0.0.1")
    )

    if args.device is None:
        args.device = "cuda" if torch.cuda.is_available() else "cpu"
    if args.device == "cpu" and args.backend == "nccl":
        args.backend = "gloo"

    args.out_fname = os.path.join(
        ClusterManager.CHECKPOINT_DIR,
        f"{args.tag}out_r{args.rank}_n{args.world_size}.csv",
    )
    args.cpu_comm = (
        args.backend == "gloo" and not args.push_sum and not args.all_reduce
    ) or args.device == "cpu"
    args.comm_device = torch.device("cpu" if args.cpu_comm else "cuda")

    args.lr_schedule = pairs_to_dict(
        args.schedule, [30, 0.1, 60, 0.1, 80, 0.1]
    )
    del args.schedule
    args.ppi_schedule = pairs_to_dict(args.peers_per_itr_schedule, [0, 1])
    del args.peers_per_itr_schedule
    assert 0 in args.ppi_schedule

    if args.all_reduce:
        assert args.graph_type == -1

    if args.network_interface_type != "auto" and args.device == "cuda":
        try:
            pin_comm_env(args.backend, args.network_interface_type)
        except (RuntimeError, AssertionError):
            pass  # fabric autodetect is best-effort outside SLURM

    os.environ["MASTER_ADDR"] = args.master_addr
    os.environ.setdefault("MASTER_PORT", args.master_port)
    if args.world_size > 1 and not dist.is_initialized():
        dist.init_process_group(
            backend=args.backend, world_size=args.world_size, rank=args.rank
        )

    args.graph, args.mixing = None, None
    graph_class = GRAPH_TOPOLOGIES[args.graph_type]
    if graph_class and args.world_size > 1:
        # barrier forces eager communicator creation in every rank at the
        # same time (reference gossip_sgd.py:678-682)
        dist.barrier()
        args.graph = graph_class(
            args.rank, args.world_size, peers_per_itr=args.ppi_schedule[0]
        )
    mixing_class = MIXING_STRATEGIES[args.mixing_strategy]
    if mixing_class and args.graph:
        args.mixing = mixing_class(args.graph, args.comm_device)
    return args


# --------------------------------------------------------------------- data

class SyntheticImageNet(torch.utils.data.Dataset):
    """Random images + labels of the ImageNet shape (no downloads here).
    Deterministic per index so epochs are reproducible."""

    def __init__(self, n, image_size=224, num_classes=1000, seed=0):
        self.n = n
        self.image_size = image_size
        self.num_classes = num_classes
        self.seed = seed

    def __len__(self):
        return self.n

    def __getitem__(self, idx):
        g = torch.Generator().manual_seed(self.seed + idx)
        x = torch.randn(3, self.image_size, self.image_size, generator=g)
        y = torch.randint(0, self.num_classes, (1,), generator=g).item()
        return x, y


def make_dataloader(args, train=True):
    """Distributed loaders; synthetic by default, ImageFolder layout when
    a dataset directory is given (reference gossip_sgd.py:539-583)."""
    if args.dataset == "imagefolder":
        from torchvision import datasets, transforms  # optional dependency

        normalize = transforms.Normalize(
            mean=[0.485, 0.456, 0.406], std=[0.229, 0.224, 0.225]
        )
        split = "train" if train else "val"
        tfm = (
            transforms.Compose([
                transforms.RandomResizedCrop(args.image_size),
                transforms.RandomHorizontalFlip(),
                transforms.ToTensor(), normalize,
            ]) if train else
            transforms.Compose([
                transforms.Resize(256), transforms.CenterCrop(args.image_size),
                transforms.ToTensor(), normalize,
            ])
        )
        dataset = datasets.ImageFolder(
            os.path.join(args.dataset_dir, split), tfm
        )
    else:
        n = args.synthetic_size if train else max(args.batch_size * 2, 64)
        dataset = SyntheticImageNet(
            n * max(args.world_size, 1), args.image_size, args.num_classes,
            seed=(0 if train else 10_000_000),
        )

    if train:
        sampler = torch.utils.data.distributed.DistributedSampler(
            dataset=dataset, num_replicas=args.world_size, rank=args.rank
        )
        loader = torch.utils.data.DataLoader(
            dataset, batch_size=args.batch_size, shuffle=False,
            num_workers=args.num_dataloader_workers,
            pin_memory=(args.device == "cuda"), sampler=sampler,
            drop_last=True,  # fixed step shape (graph capture, lr scaling)
        )
        return loader, sampler
    return torch.utils.data.DataLoader(
        dataset, batch_size=args.batch_size, shuffle=False,
        num_workers=args.num_dataloader_workers,
        pin_memory=(args.device == "cuda"),
    )


# ---------------------------------------------------------------- training

def accuracy(output, target, topk=(1,)):
    """Precision@k (reference gossip_sgd.py:474-488)."""
    with torch.no_grad():
        maxk = max(topk)
        batch_size = target.size(0)
        _, pred = output.topk(maxk, 1, True, True)
        pred = pred.t()
        correct = pred.eq(target.view(1, -1).expand_as(pred))
        return [
            correct[:k].reshape(-1).float().sum(0, keepdim=True)
            .mul_(100.0 / batch_size)
            for k in topk
        ]


def update_state(state, update_dict):
    for key, v in update_dict.items():
        state[key] = copy.deepcopy(v)


def update_peers_per_itr(args, model, epoch):
    """Apply the peers-per-iteration schedule (reference
    gossip_sgd.py:497-505)."""
    ppi, e_max = None, -1
    for e, v in args.ppi_schedule.items():
        if e_max <= e <= epoch:
            e_max, ppi = e, v
    if ppi is not None:
        model.update_gossiper("peers_per_itr", ppi)


def update_learning_rate(args, optimizer, epoch, itr=None,
                         itr_per_epoch=None, scale=1):
    """Linear warmup to the scaled reference lr over 5 epochs, then step
    decay (reference gossip_sgd.py:508-536)."""
    target_lr = args.lr * args.batch_size * scale * args.world_size / 256

    if args.warmup and epoch < 5:
        if target_lr <= args.lr:
            lr = target_lr
        else:
            assert itr is not None and itr_per_epoch is not None
            count = epoch * itr_per_epoch + itr + 1
            incr = (target_lr - args.lr) * (count / (5 * itr_per_epoch))
            lr = args.lr + incr
    else:
        lr = target_lr
        for e, factor in args.lr_schedule.items():
            if epoch >= e:
                lr *= factor

    for group in optimizer.param_groups:
        group["lr"] = lr
    return lr


def make_criterion(args):
    """KLDiv against a one-hot target (reference gossip_sgd.py:192-198)."""
    core = nn.KLDivLoss(reduction="batchmean")
    log_softmax = nn.LogSoftmax(dim=1)

    def criterion(output, kl_target):
        assert kl_target.dtype != torch.int64
        return core(log_softmax(output.float()), kl_target)

    return criterion


def one_hot(target, num_classes, device):
    return torch.zeros(
        target.shape[0], num_classes, device=device
    ).scatter_(1, target.view(-1, 1), 1)


class GraphedStep:
    """hipGraph-captured fwd+bwd+FusedSGD step for the trainer.

    The gossip state machine (query/transfer, host flag logic, fused
    residual merge) runs eagerly around the replay; push-sum scalars and
    the learning rate are device-resident so replay sees live values.
    Requires lazy mixing (sync SGP/D-PSGD) — bias/de-bias are no-ops
    there, keeping the captured region pure compute.
    """

    WARMUP_STEPS = 3

    def __init__(self, args, gdp, optimizer, criterion):
        self.args = args
        self.gdp = gdp
        self.inner = gdp.module
        self.optimizer = optimizer
        self.criterion = criterion
        self.graph = None
        self.calls = 0
        self.sx = None
        self.st = None
        self.out = None
        self.loss = None

    def _compute(self):
        with torch.autocast(device_type="cuda", dtype=torch.bfloat16):
            out = self.inner(self.sx)
        loss = self.criterion(out, self.st)
        loss.backward()
        self.optimizer.step()
        self.optimizer.zero_grad()
        # return detached handles: a live autograd graph held by the
        # caller across the capture call keeps stale AccumulateGrad
        # nodes alive and segfaults stream capture
        return out.detach(), loss.detach()

    def __call__(self, batch, kl_target):
        if self.sx is not None and batch.shape != self.sx.shape:
            # ragged batch (shouldn't happen with drop_last, but never
            # crash on it): run this one eagerly outside the graph
            with torch.autocast(device_type="cuda", dtype=torch.bfloat16):
                out = self.inner(batch)
            loss = self.criterion(out, kl_target)
            loss.backward()
            self.optimizer.step()
            self.optimizer.zero_grad()
            if self.gdp.distributed:
                self.gdp._query_gossip_queue(non_blocking=self.gdp.asynch)
                self.gdp.transfer_params()
            return out.detach(), loss.detach()
        if self.sx is None:
            self.sx = batch.clone()
            self.st = kl_target.clone()
        else:
            self.sx.copy_(batch, non_blocking=True)
            self.st.copy_(kl_target, non_blocking=True)
        if self.graph is None:
            if self.calls < self.WARMUP_STEPS:
                self.calls += 1
                out, loss = self._compute()
            else:
                torch.cuda.synchronize()
                self.graph = torch.cuda.CUDAGraph()
                with torch.cuda.graph(
                    self.graph, capture_error_mode="thread_local"
                ):
                    out, loss = self._compute()
                self.out, self.loss = out, loss
        else:
            self.graph.replay()
            out, loss = self.out, self.loss
        if self.gdp.distributed:
            self.gdp._query_gossip_queue(non_blocking=self.gdp.asynch)
            self.gdp.transfer_params()
        return out, loss


def train_epoch(args, log, model, criterion, optimizer, batch_meter,
                data_meter, nn_meter, loader, epoch, start_itr, begin_time,
                num_itr_ignore, graphed_step=None):
    losses = Meter(ptag="Loss")
    top1 = Meter(ptag="Prec@1")
    top5 = Meter(ptag="Prec@5")
    model.train()
    device = torch.device(args.device)
    is_gossip = not args.all_reduce

    it = iter(loader)
    # mid-epoch resume: fast-forward the sampler (reference
    # gossip_sgd.py:356-364)
    for i in range(start_itr):
        try:
            next(it)
        except StopIteration:
            log.info(f"Loader spoof error attempt {i}/{len(loader)}")
            return

    batch_time = time.time()
    i = start_itr - 1
    for i, (batch, target) in enumerate(it, start=start_itr):
        batch = batch.to(device, non_blocking=True)
        target = target.to(device, non_blocking=True)
        if args.device == "cuda":
            batch = batch.to(memory_format=torch.channels_last)
        kl_target = one_hot(target, args.num_classes, device)
        if num_itr_ignore == 0:
            data_meter.update(time.time() - batch_time)

        nn_time = time.time()
        if graphed_step is not None:
            if i % 100 == 0:
                update_learning_rate(args, optimizer, epoch, itr=i,
                                     itr_per_epoch=len(loader))
                optimizer.sync_lr()
            output, loss = graphed_step(batch, kl_target)
        else:
            with torch.autocast(
                device_type=args.device, dtype=torch.bfloat16,
                enabled=(args.device == "cuda"),
            ):
                output = model(batch)
            loss = criterion(output, kl_target)
            loss.backward()
            if i % 100 == 0:
                update_learning_rate(args, optimizer, epoch, itr=i,
                                     itr_per_epoch=len(loader))
            optimizer.step()
            optimizer.zero_grad()
            if is_gossip and not args.overlap:
                model.transfer_params()
        if num_itr_ignore == 0:
            nn_meter.update(time.time() - nn_time)
            batch_meter.update(time.time() - batch_time)
        batch_time = time.time()

        prec1, prec5 = accuracy(output.float(), target, topk=(1, 5))
        losses.update(loss.item(), batch.size(0))
        top1.update(prec1.item(), batch.size(0))
        top5.update(prec5.item(), batch.size(0))
        if i % args.print_freq == 0:
            write_train_row(args, epoch, i, batch_meter, nn_meter,
                            data_meter, losses, top1, top5)
        if num_itr_ignore > 0:
            num_itr_ignore -= 1
        if (args.num_iterations_per_training_epoch != -1
                and i + 1 == args.num_iterations_per_training_epoch):
            break

    write_train_row(args, epoch, i, batch_meter, nn_meter, data_meter,
                    losses, top1, top5)


def write_train_row(args, epoch, itr, bt, nt, dt, losses, top1, top5):
    with open(args.out_fname, "+a") as f:
        print(
            f"{epoch},{itr},{bt},{nt},{dt},"
            f"{losses.val:.4f},{losses.avg:.4f},"
            f"{top1.val:.3f},{top1.avg:.3f},"
            f"{top5.val:.3f},{top5.avg:.3f},-1",
            file=f,
        )


def validate(args, log, val_loader, model, criterion):
    losses = Meter(ptag="Loss")
    top1 = Meter(ptag="Prec@1")
    top5 = Meter(ptag="Prec@5")
    model.eval()
    device = torch.device(args.device)
    with torch.no_grad():
        for features, target in val_loader:
            features = features.to(device, non_blocking=True)
            target = target.to(device, non_blocking=True)
            if args.device == "cuda":
                features = features.to(memory_format=torch.channels_last)
            kl_target = one_hot(target, args.num_classes, device)
            with torch.autocast(
                device_type=args.device, dtype=torch.bfloat16,
                enabled=(args.device == "cuda"),
            ):
                output = model(features)
            loss = criterion(output, kl_target)
            prec1, prec5 = accuracy(output.float(), target, topk=(1, 5))
            losses.update(loss.item(), features.size(0))
            top1.update(prec1.item(), features.size(0))
            top5.update(prec5.item(), features.size(0))
    log.info(f" * Prec@1 {top1.avg:.3f} Prec@5 {top5.avg:.3f}")
    return top1.avg


def init_model(args):
    """ResNet with the 'ImageNet in 1hr' init (reference
    gossip_sgd.py:693-707); our build_resnet applies zero-gamma + fc init
    already."""
    model = build_resnet(
        args.model, num_classes=args.num_classes,
        zero_init_residual=True, norm=args.norm,
        conv_impl=args.conv_impl,
    )
    model = model.to(args.device)
    if args.device == "cuda":
        model = model.to(memory_format=torch.channels_last)
    return model


def main(argv=None):
    args = parse_args(argv)
    log = make_logger(args.rank, args.verbose)
    log.info(f"args: {args}")
    log.info(socket.gethostname())

    torch.manual_seed(args.seed)
    if args.device == "cuda":
        torch.cuda.manual_seed(args.seed)
        torch.backends.cudnn.benchmark = True

    model = init_model(args)
    if args.all_reduce:
        if args.world_size > 1:
            model = torch.nn.parallel.DistributedDataParallel(model)
        optimizer = torch.optim.SGD(
            model.parameters(), lr=args.lr, momentum=args.momentum,
            weight_decay=args.weight_decay, nesterov=args.nesterov,
        )
    else:
        model = GossipDataParallel(
            model,
            graph=args.graph,
            mixing=args.mixing,
            comm_device=args.comm_device,
            push_sum=args.push_sum,
            overlap=args.overlap,
            synch_freq=args.synch_freq,
            verbose=args.verbose,
            use_streams=not args.no_cuda_streams,
            rank=args.rank,
            world_size=args.world_size,
            working_dtype=(
                torch.bfloat16 if args.fused_sgd and (
                    args.master_weights == "on"
                    or (args.master_weights == "auto"
                        and args.device == "cuda")
                ) else None
            ),
        )
        if args.fused_sgd:
            optimizer = FusedSGD(
                model.flatp, lr=args.lr, momentum=args.momentum,
                weight_decay=args.weight_decay, nesterov=args.nesterov,
                steal_grads=(
                    args.steal_grads == "on"
                    or (args.steal_grads == "auto"
                        and args.device == "cuda")
                ),
            )
        else:
            optimizer = torch.optim.SGD(
                model.parameters(), lr=args.lr, momentum=args.momentum,
                weight_decay=args.weight_decay, nesterov=args.nesterov,
            )
    criterion = make_criterion(args)
    optimizer.zero_grad()

    state = {}
    update_state(state, {
        "epoch": 0, "itr": 0, "best_prec1": 0, "is_best": True,
        "state_dict": model.state_dict(),
        "optimizer": optimizer.state_dict(),
        "elapsed_time": 0,
        "batch_meter": Meter(ptag="Time").state_dict(),
        "data_meter": Meter(ptag="Data").state_dict(),
        "nn_meter": Meter(ptag="Forward/Backward").state_dict(),
    })

    cmanager = ClusterManager(
        rank=args.rank, world_size=args.world_size, model_tag=args.tag,
        state=state, all_workers=args.checkpoint_all,
    )

    if args.resume and os.path.isfile(cmanager.checkpoint_fpath):
        log.info(f"=> loading checkpoint '{cmanager.checkpoint_fpath}'")
        checkpoint = torch.load(cmanager.checkpoint_fpath,
                                weights_only=False)
        update_state(state, {
            k: checkpoint[k]
            for k in ("epoch", "itr", "best_prec1", "state_dict",
                      "optimizer", "elapsed_time", "batch_meter",
                      "data_meter", "nn_meter")
        })
        state["is_best"] = False
        model.load_state_dict(checkpoint["state_dict"])
        optimizer.load_state_dict(checkpoint["optimizer"])
        log.info(
            f"=> loaded checkpoint (epoch {checkpoint['epoch']};"
            f" itr {checkpoint['itr']})"
        )

    batch_meter = Meter(init_dict=state["batch_meter"], ptag="Time")
    data_meter = Meter(init_dict=state["data_meter"], ptag="Data")
    nn_meter = Meter(init_dict=state["nn_meter"], ptag="Forward/Backward")

    if not os.path.exists(args.out_fname):
        with open(args.out_fname, "w") as f:
            print(
                "BEGIN-TRAINING\n"
                f"World-Size,{args.world_size}\n"
                f"Num-DLWorkers,{args.num_dataloader_workers}\n"
                f"Batch-Size,{args.batch_size}\n"
                "Epoch,itr,BT(s),avg:BT(s),std:BT(s),"
                "NT(s),avg:NT(s),std:NT(s),"
                "DT(s),avg:DT(s),std:DT(s),"
                "Loss,avg:Loss,Prec@1,avg:Prec@1,Prec@5,avg:Prec@5,val",
                file=f,
            )

    loader, sampler = make_dataloader(args, train=True)
    val_loader = None
    if not args.train_fast:
        val_loader = make_dataloader(args, train=False)

    start_itr = state["itr"]
    start_epoch = state["epoch"]
    elapsed_time = state["elapsed_time"]
    begin_time = time.time() - elapsed_time
    best_val_prec1 = state.get("best_prec1", 0)
    is_gossip = not args.all_reduce

    graphed_step = None
    if (
        args.hip_graph and args.device == "cuda" and is_gossip
        and args.fused_sgd and not args.overlap
        and getattr(model, "lazy_mixing", False) is not False
    ):
        graphed_step = GraphedStep(args, model, optimizer, criterion)
        log.info("hipGraph-captured train step enabled")

    for epoch in range(start_epoch, args.num_epochs):
        sampler.set_epoch(epoch + args.seed * 90)
        if is_gossip:
            update_peers_per_itr(args, model, epoch)
            model.block()
        train_epoch(args, log, model, criterion, optimizer, batch_meter,
                    data_meter, nn_meter, loader, epoch, start_itr,
                    begin_time, args.num_itr_ignore,
                    graphed_step=graphed_step)
        start_itr = 0
        if not args.train_fast:
            elapsed_time = time.time() - begin_time
            update_state(state, {
                "epoch": epoch + 1, "itr": start_itr, "is_best": False,
                "state_dict": model.state_dict(),
                "optimizer": optimizer.state_dict(),
                "elapsed_time": elapsed_time,
                "batch_meter": batch_meter.state_dict(),
                "data_meter": data_meter.state_dict(),
                "nn_meter": nn_meter.state_dict(),
            })
            prec1 = validate(args, log, val_loader, model, criterion)
            with open(args.out_fname, "+a") as f:
                print(
                    f"{epoch},-1,{batch_meter},{nn_meter},{data_meter},"
                    f"-1,-1,-1,-1,-1,-1,{prec1}",
                    file=f,
                )
            if prec1 > best_val_prec1:
                update_state(state, {"is_best": True,
                                     "best_prec1": prec1})
                best_val_prec1 = prec1
            epoch_id = epoch if not args.overwrite_checkpoints else None
            cmanager.save_checkpoint(
                epoch_id, requeue_on_signal=(epoch != args.num_epochs - 1)
            )

    if args.train_fast:
        val_loader = make_dataloader(args, train=False)
        prec1 = validate(args, log, val_loader, model, criterion)
        log.info(f"Test accuracy: {prec1}")

    log.info(f"elapsed_time {elapsed_time}")
    if is_gossip and hasattr(model, "shutdown"):
        model.shutdown()
    if args.world_size > 1:
        dist.barrier()
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
