"""Distributed-training worker: one process per GPU rank.

Parity: /root/reference/maggy/core/executors/torch_dist_executor.py:63-422 —
rank env setup, process-group init, module wrapper construction with the
user's ctor signature preserved ("pass the class, not the instance"),
DataLoader/optimizer patching, barrier before exit.  The reference's
EXEC_CONFIG round-trip to learn the master address disappears: all ranks
are on one node, MASTER_ADDR=127.0.0.1.
"""
import builtins
import os
import time
import traceback

import torch

from maggy_amd import util
from maggy_amd.core import messages as M
from maggy_amd.core.reporter import Reporter
from maggy_amd.core.shm import MetricRing


class DDPModuleWrapper:
    """Class factory: wraps a user module class so that instantiating it
    inside train_fn yields a device-placed, distribution-wrapped module
    (parity: MaggyDDPModuleWrapper, patching/modules.py:38-65).

    zero_lvl 0/1: DDP (bucketed all-reduce over xGMI).  zero_lvl 2: no DDP
    — gradients are reduce-scattered to their shard owner inside the
    patched ZeRO optimizer's step(), so the module is only device-placed
    and its initial parameters broadcast from rank 0."""

    @classmethod
    def build(cls, module_cls, bucket_cap_mb=None, zero_lvl=0):
        class _Wrapped(module_cls):
            def __new__(wcls, *args, **kwargs):
                inner = module_cls(*args, **kwargs)
                if torch.cuda.is_available():
                    inner = inner.cuda()
                if zero_lvl >= 2:
                    import torch.distributed as dist

                    from maggy_amd.parallel.zero import (
                        broadcast_module_params,
                    )

                    if dist.is_initialized() and dist.get_world_size() > 1:
                        broadcast_module_params(inner)
                    return inner
                from maggy_amd.parallel.dist import wrap_ddp

                return wrap_ddp(inner, bucket_cap_mb=bucket_cap_mb)

        _Wrapped.__name__ = "Maggy" + module_cls.__name__
        return _Wrapped


def _patch_torch(zero_lvl):
    """Monkey-patch DataLoader (+ optimizers under ZeRO) for distribution
    transparency (parity torch_dist_executor.py:408-422)."""
    import torch.utils.data as tud

    from maggy_amd.parallel.data import MaggyDataLoader

    originals = {"DataLoader": tud.DataLoader}
    tud.DataLoader = MaggyDataLoader
    if zero_lvl and zero_lvl > 0:
        import torch.optim as topt

        from maggy_amd.parallel.zero import ZeroFusedAdam, ZeroFusedSGD

        if zero_lvl >= 2:
            # level 2: gradient reduction is sharded too (no DDP all-reduce)
            class _Adam(ZeroFusedAdam):
                def __init__(self, params, **kw):
                    kw.setdefault("grad_shard", True)
                    super().__init__(params, **kw)

            class _SGD(ZeroFusedSGD):
                def __init__(self, params, **kw):
                    kw.setdefault("grad_shard", True)
                    super().__init__(params, **kw)
        else:
            _Adam, _SGD = ZeroFusedAdam, ZeroFusedSGD
        originals["Adam"] = topt.Adam
        originals["AdamW"] = topt.AdamW
        originals["SGD"] = topt.SGD
        topt.Adam = _Adam
        topt.AdamW = _Adam
        topt.SGD = _SGD
    return originals


def _unpatch_torch(originals):
    import torch.optim as topt
    import torch.utils.data as tud

    tud.DataLoader = originals["DataLoader"]
    for name in ("Adam", "AdamW", "SGD"):
        if name in originals:
            setattr(topt, name, originals[name])


def dist_worker_main(rank, world_size, gpu_id, conn, ring_name, ring_slots,
                     log_dir, payload):
    """Entry of one DP rank (spawned by TorchDistributedTrainingDriver)."""
    # pin this process to one GPU; LOCAL_RANK stays 0 (reference idiom,
    # torch_dist_executor.py:131)
    if gpu_id is not None:
        os.environ["HIP_VISIBLE_DEVICES"] = str(gpu_id)
        os.environ["CUDA_VISIBLE_DEVICES"] = str(gpu_id)
    os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    os.environ["LOCAL_RANK"] = "0"
    os.environ["MASTER_ADDR"] = payload.get("master_addr", "127.0.0.1")
    os.environ["MASTER_PORT"] = str(payload.get("master_port", 29500))

    import torch.distributed as dist

    ring = MetricRing(name=ring_name, slots=ring_slots)
    reporter = Reporter(
        ring=ring,
        log_file=os.path.join(log_dir, "executor_{}.log".format(rank)),
        worker_id=rank,
    )
    real_print = builtins.print

    def maggy_print(*args, **kwargs):
        real_print(*args, **kwargs)
        reporter.log(" ".join(str(x) for x in args), True)

    originals = None
    try:
        conn.send((M.REG, rank, os.getpid()))
        backend = "nccl" if torch.cuda.is_available() else "gloo"
        from maggy_amd.parallel.dist import init_process_group

        init_process_group(backend=backend,
                           timeout_s=payload.get("timeout_s", 120))
        if torch.cuda.is_available():
            torch.cuda.set_device(0)

        module_cls = payload["module"]
        zero_lvl = payload.get("zero_lvl", 0)
        wrapped = None
        if module_cls is not None:
            wrapped = DDPModuleWrapper.build(
                module_cls, bucket_cap_mb=payload.get("bucket_cap_mb"),
                zero_lvl=zero_lvl)
        originals = _patch_torch(zero_lvl)

        trial_dir = os.path.join(log_dir, "dist_run")
        os.makedirs(trial_dir, exist_ok=True)
        # synthetic 16-hex trial id per rank (tags the metric ring records)
        reporter.set_trial_id("d157000000000{:03x}".format(rank % 4096))
        reporter.init_logger(
            os.path.join(trial_dir, "output_rank{}.log".format(rank)))

        train_fn = payload["train_fn"]
        hparams = payload.get("hparams", {}) or {}
        builtins.print = maggy_print
        start = time.time()
        try:
            kwargs = util.build_train_kwargs(
                train_fn,
                model=None,
                dataset=payload.get("dataset"),
                hparams=hparams,
                reporter=reporter,
                extra={"module": wrapped,
                       "test_set": payload.get("test_set"),
                       "rank": rank,
                       "world_size": world_size},
            )
            if payload.get("mixed_precision"):
                # the reference's mixed_precision flag mapped to fairscale
                # FSDP fp16 (modules.py:75-97); MI355X-native: bf16
                # autocast around the whole training function
                device_type = "cuda" if torch.cuda.is_available() else "cpu"
                with torch.autocast(device_type=device_type,
                                    dtype=torch.bfloat16):
                    retval = train_fn(**kwargs)
            else:
                retval = train_fn(**kwargs)
            if rank == 0 and retval is not None:
                util.handle_return_val(
                    retval, trial_dir,
                    payload.get("optimization_key", "Metric"))
            if isinstance(retval, dict):
                opt_val = retval.get(payload.get("optimization_key",
                                                 "Metric"))
            else:
                opt_val = retval
            if dist.is_initialized():
                dist.barrier()  # avoid RCCL teardown crash (reference :168)
            conn.send((M.FINAL, rank, "dist_run", opt_val,
                       time.time() - start, False, reporter.logs))
        except Exception:
            tb = traceback.format_exc()
            reporter.log(tb, False)
            conn.send((M.ERROR, rank, "dist_run", tb))
        finally:
            builtins.print = real_print
            if dist.is_initialized():
                dist.destroy_process_group()
    finally:
        if originals is not None:
            _unpatch_torch(originals)
        builtins.print = real_print
        reporter.close_logger()
        ring.close()
        conn.close()
