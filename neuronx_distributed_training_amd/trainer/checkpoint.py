"""Sharded checkpoint engine (reference nxd.save_checkpoint /
load_checkpoint contract, nlp_overrides.py:535-639 + dir layout from
nnm_model_ckpt_to_nxdt_model_ckpt_converter.py:77-113).

On-disk layout::

    <dir>/<tag>.ckpt/
        model/dp_rank_00_tp_rank_XX_pp_rank_XX.pt    # model shard per rank
        optim/dp_rank_XX_tp_rank_XX_pp_rank_XX.pt    # ZeRO-1 shard per rank
        user_content.pt                               # loop state, config
        done                                          # commit marker

Model shards are written once per (tp, pp) coordinate (dp rank 0);
optimizer shards are per-DP-rank (ZeRO-1 state is DP-sharded). Async save
runs in a background thread over CPU copies (reference async_checkpointing
semantics); keep-top-k prunes oldest step tags.
"""

from __future__ import annotations

import os
import re
import shutil
import threading
from typing import Dict, List, Optional

import torch
import torch.distributed as dist

from ..parallel import state as ps


def _rank_tags():
    return (
        ps.get_data_parallel_rank(),
        ps.get_tensor_model_parallel_rank(),
        ps.get_pipeline_model_parallel_rank(),
    )


def _model_shard_name() -> str:
    _, tp, pp = _rank_tags()
    return f"dp_rank_00_tp_rank_{tp:02d}_pp_rank_{pp:02d}.pt"


def _optim_shard_name() -> str:
    dp, tp, pp = _rank_tags()
    return f"dp_rank_{dp:02d}_tp_rank_{tp:02d}_pp_rank_{pp:02d}.pt"


def _cpu_copy_inner(obj, pin: bool):
    if torch.is_tensor(obj):
        if obj.is_cuda and pin:
            # pinned staging → true async D2H on the side stream (pageable
            # destinations force a synchronous copy in HIP)
            dst = torch.empty(
                obj.shape, dtype=obj.dtype, device="cpu", pin_memory=True
            )
            dst.copy_(obj.detach(), non_blocking=True)
            return dst
        return obj.detach().to("cpu", non_blocking=True)
    if isinstance(obj, dict):
        return {k: _cpu_copy_inner(v, pin) for k, v in obj.items()}
    if isinstance(obj, (list, tuple)):
        t = type(obj)
        return t(_cpu_copy_inner(v, pin) for v in obj)
    return obj


_COPY_STREAM = None


def _cpu_copy(obj):
    """D2H snapshot. On GPU the copies run on a dedicated side stream into
    pinned buffers and this returns WITHOUT waiting — the returned event
    gates the writer (reference async_checkpointing: the training loop
    must not stall for checkpoint IO, known_issues.rst:55-85). Returns
    (cpu_obj, event_or_None)."""
    global _COPY_STREAM
    if not (torch.cuda.is_available() and torch.cuda.is_initialized()):
        return _cpu_copy_inner(obj, pin=False), None
    if _COPY_STREAM is None:
        _COPY_STREAM = torch.cuda.Stream()
    _COPY_STREAM.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(_COPY_STREAM):
        out = _cpu_copy_inner(obj, pin=True)
        ev = torch.cuda.Event()
        ev.record(_COPY_STREAM)
    # GPU-side ordering only: later kernels (the next optimizer step
    # mutates these params in place) queue behind the copies, but the
    # HOST thread returns immediately instead of torch.cuda.synchronize()
    torch.cuda.current_stream().wait_event(ev)
    return out, ev


def _remote_fs(path: str):
    """fsspec filesystem for scheme-qualified paths (s3://, gcs://,
    memory:// in tests) — the reference's "dump a checkpoint to S3"
    capability (features.rst checkpointing list). None for local paths.
    Remote checkpoints skip atomic-rename and top-k pruning."""
    if "://" not in str(path) or str(path).startswith("file://"):
        return None
    import fsspec

    return fsspec.filesystem(str(path).split("://", 1)[0])


def _join(root: str, fs, *parts: str) -> str:
    if fs is not None:
        return "/".join([root.rstrip("/"), *parts])
    return os.path.join(root, *parts)


def _touch_done(root: str, fs):
    if fs is not None:
        with fs.open(_join(root, fs, "done"), "wb") as f:
            f.write(b"")
    else:
        open(os.path.join(root, "done"), "w").close()


class CheckpointIO:
    def __init__(self, async_save: bool = False, save_bf16: bool = False,
                 writer_process: bool = False):
        self.async_save = async_save
        # reference exp_manager `save_bf16`: cast fp32 model tensors to
        # bf16 on save (halves shard size; optimizer masters stay fp32)
        self.save_bf16 = save_bf16
        # fork a writer process per save so torch.save's pickling (which
        # holds the GIL) leaves the training process entirely
        self.writer_process = writer_process
        self._pending: List[threading.Thread] = []

    # ---- save ----
    def save(
        self,
        ckpt_dir: str,
        tag: str,
        module,
        user_content: Dict,
        keep_top_k: int = 0,
    ):
        fs = _remote_fs(ckpt_dir)
        if fs is not None:
            root = f"{ckpt_dir.rstrip('/')}/{tag}.ckpt"
            fs.makedirs(f"{root}/model", exist_ok=True)
            fs.makedirs(f"{root}/optim", exist_ok=True)
        else:
            root = os.path.join(ckpt_dir, f"{tag}.ckpt")
            os.makedirs(os.path.join(root, "model"), exist_ok=True)
            os.makedirs(os.path.join(root, "optim"), exist_ok=True)

        dp, tp, pp = _rank_tags()
        work = []
        events = []
        if dp == 0 and ps.get_context_model_parallel_rank() == 0:
            msd, ev = _cpu_copy(module.model.state_dict())
            events.append(ev)
            if self.save_bf16:
                # the cast reads the pinned copies — it must run after the
                # D2H completes, so it moves into the writer
                def _cast(sd=msd):
                    return {
                        k: (v.to(torch.bfloat16)
                            if torch.is_tensor(v) and v.dtype == torch.float32
                            else v)
                        for k, v in sd.items()
                    }
                msd = _cast
            work.append(
                (_join(root, fs, "model", _model_shard_name()), msd)
            )
        if module.optimizer is not None and ps.get_context_model_parallel_rank() == 0:
            osd, ev = _cpu_copy(module.optimizer.state_dict())
            events.append(ev)
            work.append(
                (_join(root, fs, "optim", _optim_shard_name()), osd)
            )
        is_global_zero = (not dist.is_initialized()) or dist.get_rank() == 0
        if is_global_zero:
            uc = dict(user_content)
            if module.scheduler is not None:
                uc["scheduler"] = module.scheduler.state_dict()
            work.append((_join(root, fs, "user_content.pt"), uc))

        # forking with a live HIP runtime deadlocks the child (allocator
        # locks held by threads that don't exist post-fork — observed
        # hanging on MI355X, r2); the writer process is therefore only
        # used before CUDA init (CPU jobs / tests). GPU saves rely on the
        # pinned side-stream staging + thread writer, which never blocks
        # the training thread.
        use_fork = (self.writer_process and fs is None
                    and hasattr(os, "fork")
                    and not torch.cuda.is_initialized())

        def _write():
            for ev in events:
                if ev is not None:
                    ev.synchronize()  # pinned copies durable before save
            items = [(p, o() if callable(o) else o) for p, o in work]

            def _dump():
                for path, obj in items:
                    if fs is not None:
                        with fs.open(path, "wb") as f:
                            torch.save(obj, f)
                    else:
                        tmp = path + ".tmp"
                        torch.save(obj, tmp)
                        os.replace(tmp, path)

            if use_fork:
                # reference async semantics: the saver is a separate
                # PROCESS (nlp_overrides.py:618-627) — serialization cost
                # (pickling holds the GIL) leaves the training process.
                # The forked child only touches CPU memory and files,
                # never the HIP context.
                pid = os.fork()
                if pid == 0:
                    try:
                        _dump()
                    finally:
                        os._exit(0)
                os.waitpid(pid, 0)
            else:
                _dump()

        if self.async_save:
            t = threading.Thread(target=_write, daemon=False)
            t.start()
            self._pending.append(t)
        else:
            _write()

        if dist.is_initialized():
            dist.barrier()
        if is_global_zero and not self.async_save:
            # sync mode: the barrier above already guarantees every rank's
            # (synchronous) writes are durable
            _touch_done(root, fs)
            if keep_top_k and fs is None:
                self._prune(ckpt_dir, keep_top_k)
        elif is_global_zero and self.async_save:
            # async mode: OTHER ranks' writer threads may still be in
            # flight after the barrier — the commit thread must not mark
            # the checkpoint complete until every expected shard file
            # exists (all writes land via tmp+atomic-rename, so presence
            # implies a complete file). Otherwise a crash in that window
            # leaves a 'done'-marked checkpoint with missing shards that
            # find_latest_checkpoint would resume from.
            expected = self._expected_shards(
                root, fs, has_optim=module.optimizer is not None
            )

            def _commit(threads=list(self._pending), root=root, k=keep_top_k,
                        d=ckpt_dir, expected=expected):
                for t in threads:
                    t.join()
                self._wait_for_shards(expected, fs)
                _touch_done(root, fs)
                if k and fs is None:
                    self._prune(d, k)
            tc = threading.Thread(target=_commit, daemon=False)
            tc.start()
            self._pending = [tc]

    @staticmethod
    def _expected_shards(root: str, fs, has_optim: bool) -> List[str]:
        """Every shard path the full job writes for this tag (global-rank-0
        view): one model shard per (tp, pp) coordinate and, when an
        optimizer is attached, one optim shard per (dp, tp, pp)."""
        tp_w = ps.get_tensor_model_parallel_world_size()
        pp_w = ps.get_pipeline_model_parallel_world_size()
        dp_w = ps.get_data_parallel_world_size()
        paths = []
        for pp in range(pp_w):
            for tp in range(tp_w):
                paths.append(_join(
                    root, fs, "model",
                    f"dp_rank_00_tp_rank_{tp:02d}_pp_rank_{pp:02d}.pt"))
                if has_optim:
                    for dp in range(dp_w):
                        paths.append(_join(
                            root, fs, "optim",
                            f"dp_rank_{dp:02d}_tp_rank_{tp:02d}"
                            f"_pp_rank_{pp:02d}.pt"))
        return paths

    @staticmethod
    def _wait_for_shards(paths: List[str], fs, timeout_s: float = 900.0):
        import time

        def _exists(p):
            return fs.exists(p) if fs is not None else os.path.exists(p)

        deadline = time.monotonic() + timeout_s
        missing = list(paths)
        while missing:
            missing = [p for p in missing if not _exists(p)]
            if not missing:
                return
            if time.monotonic() > deadline:
                raise TimeoutError(
                    f"checkpoint commit: {len(missing)} shard(s) never "
                    f"appeared (first: {missing[0]}); refusing to mark done"
                )
            time.sleep(0.2)

    def finalize(self):
        for t in self._pending:
            t.join()
        self._pending = []

    def _prune(self, ckpt_dir: str, keep: int):
        tags = []
        for name in os.listdir(ckpt_dir):
            m = re.match(r"step=(\d+).*\.ckpt$", name)
            if m and os.path.exists(os.path.join(ckpt_dir, name, "done")):
                tags.append((int(m.group(1)), name))
        tags.sort()
        for _, name in tags[:-keep] if keep > 0 else []:
            shutil.rmtree(os.path.join(ckpt_dir, name), ignore_errors=True)

    # ---- load ----
    def load(self, path: str, module, weight_init_only: bool = False,
             broadcast_over_dp: bool = False) -> Dict:
        """path: .../<tag>.ckpt directory. Returns user_content.

        ``broadcast_over_dp``: only DP rank 0 reads the (DP-replicated)
        model shard from disk; the tensors are broadcast over the DP group
        (reference features.rst: avoids filesystem contention on network
        storage). Optimizer shards are per-rank and always read locally.
        """
        fs = _remote_fs(path)
        if broadcast_over_dp and dist.is_initialized() \
                and ps.get_data_parallel_world_size() > 1:
            return self._load_broadcast(path, module, weight_init_only, fs)

        def _ld(p):
            if fs is not None:
                with fs.open(p, "rb") as f:
                    return torch.load(f, map_location="cpu", weights_only=False)
            return torch.load(p, map_location="cpu", weights_only=False)

        def _exists(p):
            return fs.exists(p) if fs is not None else os.path.exists(p)

        sd = _ld(_join(path, fs, "model", _model_shard_name()))
        module.model.load_state_dict(sd)
        if not weight_init_only and module.optimizer is not None:
            opath = _join(path, fs, "optim", _optim_shard_name())
            if _exists(opath):
                module.optimizer.load_state_dict(_ld(opath))
        uc_path = _join(path, fs, "user_content.pt")
        uc = {}
        if _exists(uc_path):
            uc = _ld(uc_path)
            if uc.get("scheduler") and module.scheduler is not None and not weight_init_only:
                module.scheduler.load_state_dict(uc["scheduler"])
        return uc


    def _load_broadcast(self, path, module, weight_init_only, fs):
        group = ps.get_data_parallel_group()
        src_is_me = ps.get_data_parallel_rank() == 0

        def _ld(p):
            if fs is not None:
                with fs.open(p, "rb") as f:
                    return torch.load(f, map_location="cpu", weights_only=False)
            return torch.load(p, map_location="cpu", weights_only=False)

        obj = [None]
        if src_is_me:
            obj[0] = _ld(_join(path, fs, "model", _model_shard_name()))
        src = dist.get_process_group_ranks(group)[0]
        dist.broadcast_object_list(obj, src=src, group=group)
        module.model.load_state_dict(obj[0])
        if not weight_init_only and module.optimizer is not None:
            opath = _join(path, fs, "optim", _optim_shard_name())
            exists = fs.exists(opath) if fs is not None else os.path.exists(opath)
            if exists:
                module.optimizer.load_state_dict(_ld(opath))
        uc = [None]
        if src_is_me:
            p2 = _join(path, fs, "user_content.pt")
            exists = fs.exists(p2) if fs is not None else os.path.exists(p2)
            uc[0] = _ld(p2) if exists else {}
        dist.broadcast_object_list(uc, src=src, group=group)
        uc = uc[0] or {}
        if uc.get("scheduler") and module.scheduler is not None \
                and not weight_init_only:
            module.scheduler.load_state_dict(uc["scheduler"])
        return uc


def find_latest_checkpoint(ckpt_dir: str) -> Optional[str]:
    """Resume discovery: newest complete `*.ckpt` dir (reference
    exp_manager.py:370-385 semantics, by step number then mtime)."""
    if not ckpt_dir or not os.path.isdir(ckpt_dir):
        return None
    best = None
    best_key = (-1, -1.0)
    for name in os.listdir(ckpt_dir):
        p = os.path.join(ckpt_dir, name)
        if not name.endswith(".ckpt") or not os.path.isdir(p):
            continue
        if not os.path.exists(os.path.join(p, "done")):
            continue
        m = re.match(r"step=(\d+)", name)
        step = int(m.group(1)) if m else 0
        key = (step, os.path.getmtime(p))
        if key > best_key:
            best_key, best = key, p
    return best
