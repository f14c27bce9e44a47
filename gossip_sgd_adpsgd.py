#!/usr/bin/env python3
"""AD-PSGD trainer CLI (asynchronous decentralized parallel SGD).

MI355X-native re-implementation of reference gossip_sgd_adpsgd.py:70-730.
Structure matches the SGP trainer with the reference's documented deltas:
CrossEntropyLoss instead of the KLDiv/one-hot pair (reference :179),
a shared-file global iteration counter (file size = total iterations,
reference :509-523), the bilateral LR schedule driven by the *global*
epoch estimate (reference :478-506), and an epoch loop whose stopping
criterion is the global epoch (reference :269-309).  The trainer process
never joins the dist world — the comm world lives in the
BilatGossipDataParallel gossip process.
"""

import argparse
import copy
import os
import socket
import time

import torch
import torch.nn as nn

from gossip_sgd import (
    GRAPH_TOPOLOGIES,
    MIXING_STRATEGIES,
    SyntheticImageNet,
    accuracy,
    discover_rank_world,
    make_dataloader,
    pairs_to_dict,
    str2bool,
    update_state,
)
from stochastic_gradient_push_amd import BilatGossipDataParallel
from stochastic_gradient_push_amd.models import build_resnet
from stochastic_gradient_push_amd.utils import Meter, make_logger
from stochastic_gradient_push_amd.utils.cluster_manager import ClusterManager
from stochastic_gradient_push_amd.utils.nic import get_tcp_interface_name


def build_parser():
    p = argparse.ArgumentParser(description="MI355X AD-PSGD trainer")
    p.add_argument("--bilat", default="True", type=str)
    p.add_argument("--all_reduce", default="False", type=str)
    p.add_argument("--shared_fpath", default="", type=str,
                   help="shared file used as the global iteration counter")
    p.add_argument("--batch_size", default=32, type=int)
    p.add_argument("--lr", default=0.1, type=float)
    p.add_argument("--num_dataloader_workers", default=4, type=int)
    p.add_argument("--num_epochs", default=90, type=int)
    p.add_argument("--num_iterations_per_training_epoch", default=-1,
                   type=int)
    p.add_argument("--momentum", default=0.9, type=float)
    p.add_argument("--weight_decay", default=1e-4, type=float)
    p.add_argument("--nesterov", default="False", type=str)
    p.add_argument("--graph_type", default=1, type=int,
                   choices=list(GRAPH_TOPOLOGIES),
                   help="bilateral gossip needs a bipartite topology")
    p.add_argument("--mixing_strategy", default=0, type=int,
                   choices=list(MIXING_STRATEGIES))
    p.add_argument("--schedule", nargs="+", default=None, type=float)
    p.add_argument("--peers_per_itr_schedule", nargs="+", type=int)
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
    p.add_argument("--network_interface_type", default="ethernet",
                   choices=["infiniband", "ethernet", "auto"])
    p.add_argument("--num_itr_ignore", type=int, default=10)
    p.add_argument("--dataset_dir", type=str, default=None)
    # MI355X-native extensions (same as gossip_sgd.py)
    p.add_argument("--dataset", default="synthetic",
                   choices=["synthetic", "imagefolder"])
    p.add_argument("--synthetic_size", default=2048, type=int)
    p.add_argument("--model", default="resnet50", type=str)
    p.add_argument("--norm", default="fused", type=str,
                   choices=["fused", "native", "miopen"])
    p.add_argument("--num_classes", default=1000, type=int)
    p.add_argument("--device", default=None, choices=[None, "cuda", "cpu"])
    p.add_argument("--image_size", default=224, type=int)
    return p


def parse_args(argv=None):
    args = build_parser().parse_args(argv)
    ClusterManager.set_checkpoint_dir(args.checkpoint_dir)
    os.makedirs(args.checkpoint_dir, exist_ok=True)

    for flag in ("bilat", "all_reduce", "nesterov", "warmup", "resume",
                 "verbose", "train_fast", "checkpoint_all",
                 "overwrite_checkpoints"):
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
    if not args.shared_fpath:
        args.shared_fpath = os.path.join(
            ClusterManager.CHECKPOINT_DIR, f"{args.tag}global_itr.share"
        )
    # comm over pinned CPU staging unless nccl device comm is requested
    args.comm_device = torch.device(
        "cuda" if (args.backend == "nccl" and args.device == "cuda")
        else "cpu"
    )

    args.lr_schedule = pairs_to_dict(
        args.schedule, [30, 0.1, 60, 0.1, 80, 0.1]
    )
    del args.schedule
    args.ppi_schedule = pairs_to_dict(args.peers_per_itr_schedule, [0, 1])
    del args.peers_per_itr_schedule
    assert 0 in args.ppi_schedule

    args.graph_class = GRAPH_TOPOLOGIES[args.graph_type]
    args.mixing_class = MIXING_STRATEGIES[args.mixing_strategy]
    args.global_itr = None
    args.global_epoch = 0
    return args


def update_global_iteration_counter(args, log, itr_per_epoch, itr=1):
    """Append `itr` bytes to the shared file; its size IS the global
    iteration count across all agents (reference :509-523)."""
    with open(args.shared_fpath, "+a") as f:
        print("-" * itr, end="", file=f)
    args.global_itr = int(os.stat(args.shared_fpath).st_size)
    args.global_epoch = int(
        args.global_itr / itr_per_epoch / args.world_size
    )
    log.debug(
        f"global epoch estimate {args.global_epoch}, "
        f"global itr estimate {args.global_itr}"
    )


def compute_bilat_lr(args, itr_per_epoch):
    """Warmup+decay lr from the *global* progress (reference :478-506)."""
    target_lr = args.lr * args.batch_size * args.world_size / 256
    epoch = args.global_epoch
    itr_per_epoch = itr_per_epoch * args.world_size
    itr = (args.global_itr or 0) % itr_per_epoch
    if args.warmup and epoch < 5:
        if target_lr <= args.lr:
            return target_lr
        count = epoch * itr_per_epoch + itr + 1
        return args.lr + (target_lr - args.lr) * (count / (5 * itr_per_epoch))
    lr = target_lr
    for e, factor in args.lr_schedule.items():
        if epoch >= e:
            lr *= factor
    return lr


def one_hot_free_criterion():
    return nn.CrossEntropyLoss()


def train_epoch(args, log, model, criterion, optimizer, batch_meter,
                data_meter, nn_meter, loader, epoch, start_itr,
                num_itr_ignore):
    losses = Meter(ptag="Loss")
    top1 = Meter(ptag="Prec@1")
    top5 = Meter(ptag="Prec@5")
    model.train()
    device = torch.device(args.device)

    batch_time = time.time()
    i = start_itr - 1
    for i, (batch, target) in enumerate(loader, start=start_itr):
        batch = batch.to(device, non_blocking=True)
        target = target.to(device, non_blocking=True)
        if args.device == "cuda":
            batch = batch.to(memory_format=torch.channels_last)
        if num_itr_ignore == 0:
            data_meter.update(time.time() - batch_time)

        nn_time = time.time()
        with torch.autocast(
            device_type=args.device, dtype=torch.bfloat16,
            enabled=(args.device == "cuda"),
        ):
            output = model(batch)
        loss = criterion(output.float(), target)

        if i % 100 == 0:
            update_global_iteration_counter(args, log, len(loader), itr=100)
            lr = compute_bilat_lr(args, len(loader))
            model.update_lr(lr)
            for group in optimizer.param_groups:
                group["lr"] = lr
        loss.backward()  # backward hook pushes grads / pulls model
        optimizer.step()
        optimizer.zero_grad()
        if num_itr_ignore == 0:
            nn_meter.update(time.time() - nn_time)
            batch_meter.update(time.time() - batch_time)
        batch_time = time.time()

        prec1, prec5 = accuracy(output.float(), target, topk=(1, 5))
        losses.update(loss.item(), batch.size(0))
        top1.update(prec1.item(), batch.size(0))
        top5.update(prec5.item(), batch.size(0))
        if i % args.print_freq == 0:
            with open(args.out_fname, "+a") as f:
                print(
                    f"{epoch},{i},{batch_meter},{nn_meter},{data_meter},"
                    f"{losses.val:.4f},{losses.avg:.4f},"
                    f"{top1.val:.3f},{top1.avg:.3f},"
                    f"{top5.val:.3f},{top5.avg:.3f},-1",
                    file=f,
                )
        if num_itr_ignore > 0:
            num_itr_ignore -= 1
        if (args.num_iterations_per_training_epoch != -1
                and i + 1 == args.num_iterations_per_training_epoch):
            break


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
            with torch.autocast(
                device_type=args.device, dtype=torch.bfloat16,
                enabled=(args.device == "cuda"),
            ):
                output = model(features)
            loss = criterion(output.float(), target)
            prec1, prec5 = accuracy(output.float(), target, topk=(1, 5))
            losses.update(loss.item(), features.size(0))
            top1.update(prec1.item(), features.size(0))
            top5.update(prec5.item(), features.size(0))
    log.info(f" * Prec@1 {top1.avg:.3f} Prec@5 {top5.avg:.3f}")
    return top1.avg


def main(argv=None):
    args = parse_args(argv)
    log = make_logger(args.rank, args.verbose)
    log.info(f"args: {args}")
    log.info(socket.gethostname())

    torch.manual_seed(args.seed)
    if args.device == "cuda":
        torch.cuda.manual_seed(args.seed)
        torch.backends.cudnn.benchmark = True

    assert args.bilat and not args.all_reduce

    model = build_resnet(
        args.model, num_classes=args.num_classes, norm=args.norm
    ).to(args.device)
    if args.device == "cuda":
        model = model.to(memory_format=torch.channels_last)

    tcp_name = None
    if args.network_interface_type != "auto":
        try:
            tcp_name = get_tcp_interface_name(args.network_interface_type)
        except (RuntimeError, KeyError):
            tcp_name = None

    model = BilatGossipDataParallel(
        model,
        master_addr=args.master_addr,
        master_port=args.master_port,
        backend=args.backend,
        world_size=args.world_size,
        rank=args.rank,
        graph_class=args.graph_class,
        mixing_class=args.mixing_class,
        comm_device=args.comm_device,
        lr=args.lr,
        momentum=args.momentum,
        weight_decay=args.weight_decay,
        nesterov=args.nesterov,
        verbose=args.verbose,
        num_peers=args.ppi_schedule[0],
        network_interface_type=(
            args.network_interface_type
            if args.network_interface_type != "auto" else None
        ),
        tcp_interface_name=tcp_name,
    )
    criterion = one_hot_free_criterion()
    optimizer = torch.optim.SGD(
        model.parameters(), lr=args.lr, momentum=args.momentum,
        weight_decay=args.weight_decay, nesterov=args.nesterov,
    )
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
        rank=args.rank, world_size=1, model_tag=args.tag, state=state,
        all_workers=args.checkpoint_all,
    )

    if args.resume and os.path.isfile(cmanager.checkpoint_fpath):
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
        log.info(f"=> loaded checkpoint (epoch {checkpoint['epoch']})")

    batch_meter = Meter(init_dict=state["batch_meter"], ptag="Time")
    data_meter = Meter(init_dict=state["data_meter"], ptag="Data")
    nn_meter = Meter(init_dict=state["nn_meter"], ptag="Forward/Backward")

    if not args.resume or not os.path.exists(args.out_fname):
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

    model.block()
    start_itr = state["itr"]
    epoch = state["epoch"]
    elapsed_time = state["elapsed_time"]
    begin_time = time.time() - elapsed_time
    update_global_iteration_counter(args, log, len(loader), itr=0)

    while args.global_epoch < args.num_epochs and epoch < args.num_epochs:
        sampler.set_epoch(epoch + args.seed * 90)
        train_epoch(args, log, model, criterion, optimizer, batch_meter,
                    data_meter, nn_meter, loader, epoch, start_itr,
                    args.num_itr_ignore)
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
            model.disable_gossip()
            prec1 = validate(args, log, val_loader, model, criterion)
            model.enable_gossip()
            with open(args.out_fname, "+a") as f:
                print(
                    f"{epoch},-1,{batch_meter},{nn_meter},{data_meter},"
                    f"-1,-1,-1,-1,-1,-1,{prec1}",
                    file=f,
                )
            cmanager.save_checkpoint(requeue_on_signal=False)
            model.block()
        epoch += 1

    if args.train_fast:
        val_loader = make_dataloader(args, train=False)
        model.disable_gossip()
        prec1 = validate(args, log, val_loader, model, criterion)
        log.info(f"Test accuracy: {prec1}")

    log.info(f"elapsed_time {elapsed_time}")


if __name__ == "__main__":
    main()
