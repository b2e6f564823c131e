"""nnU-Net glue utilities (capability of reference
fl4health/utils/nnunet_utils.py:40-580): config enum, env/module-reload glue,
deep-supervision list<->dict conversion, segmentation helpers, the
multiprocess dataloader wrapper with explicit shutdown, loss/schedule
wrappers and logger-stream redirection. Native implementations — the nnunetv2
package is optional (import-guarded in clients/nnunet_client.py)."""
from __future__ import annotations

import functools
import importlib
import io
import logging
import os
import signal
import sys
from enum import Enum
from logging import Logger
from typing import Any, Callable, Sequence

import torch
import torch.multiprocessing as mp
import torch.nn as nn
from torch.nn.modules.loss import _Loss
from torch.optim.lr_scheduler import _LRScheduler
from torch.utils.data import DataLoader

from fl4health_amd.models.unet3d import DeepSupervisionLoss, PolyLRScheduler  # noqa: F401

log = logging.getLogger(__name__)


class NnunetConfig(Enum):
    """The possible nnunet model configs (reference nnunet_utils.py:40-66)."""

    _2D = "2d"
    _3D_FULLRES = "3d_fullres"
    _3D_LOWRES = "3d_lowres"
    _3D_CASCADE = "3d_cascade_fullres"


NNUNET_N_SPATIAL_DIMS = {
    NnunetConfig._2D: 2,
    NnunetConfig._3D_FULLRES: 3,
    NnunetConfig._3D_LOWRES: 3,
    NnunetConfig._3D_CASCADE: 3,
}

NNUNET_DEFAULT_NP = {
    NnunetConfig._2D: 8,
    NnunetConfig._3D_FULLRES: 4,
    NnunetConfig._3D_LOWRES: 8,
    NnunetConfig._3D_CASCADE: 4,
}


def use_default_signal_handlers(fn: Callable) -> Callable:
    """Restore default signal handlers around subprocess-spawning fns
    (reference :68-94): a gRPC/collective runtime overrides SIGINT/SIGTERM,
    which breaks `fork`/`spawn` children."""

    @functools.wraps(fn)
    def wrapped(*args: Any, **kwargs: Any) -> Any:
        sigint = signal.getsignal(signal.SIGINT)
        sigterm = signal.getsignal(signal.SIGTERM)
        signal.signal(signal.SIGINT, signal.SIG_DFL)
        signal.signal(signal.SIGTERM, signal.SIG_DFL)
        try:
            return fn(*args, **kwargs)
        finally:
            signal.signal(signal.SIGINT, sigint)
            signal.signal(signal.SIGTERM, sigterm)

    return wrapped


def reload_modules(packages: Sequence[str]) -> None:
    """Reload already-imported modules whose name starts with one of the
    given package names (reference :96-113); needed after changing nnunet env
    vars, which nnunetv2 reads at import time."""
    for name in sorted(list(sys.modules.keys()), key=len):
        if any(name == p or name.startswith(p + ".") for p in packages):
            mod = sys.modules.get(name)
            if mod is None:
                continue
            try:
                importlib.reload(mod)
            except Exception:  # noqa: BLE001 — some modules refuse reload; best effort
                log.debug("could not reload module %s", name)


def set_nnunet_env(verbose: bool = False, **kwargs: str) -> None:
    """Set nnunet env vars (nnUNet_raw / nnUNet_preprocessed / nnUNet_results
    ...; reference :145-165)."""
    for k, v in kwargs.items():
        os.environ[k] = str(v)
        if verbose:
            log.info("set %s=%s", k, v)


def set_nnunet_env_and_reload_modules(verbose: bool = False, **kwargs: str) -> None:
    """Set env vars then reload nnunet modules so the new paths take effect
    (reference :115-143)."""
    set_nnunet_env(verbose, **kwargs)
    reload_modules(["nnunetv2", "batchgenerators"])


# ---------------------------------------------------------------------------
# deep-supervision list <-> dict (reference :167-211)
# ---------------------------------------------------------------------------

def convert_deep_supervision_list_to_dict(
    tensor_list: list[torch.Tensor] | tuple[torch.Tensor, ...], num_spatial_dims: int
) -> dict[str, torch.Tensor]:
    """Name each deep-supervision head by its spatial resolution:
    'prediction-<i>-<XxYxZ>' with index 0 = the full-resolution head."""
    out = {}
    for i, t in enumerate(tensor_list):
        spatial = "x".join(str(s) for s in t.shape[-num_spatial_dims:])
        key = "prediction" if i == 0 else f"prediction-ds{i}-{spatial}"
        out[key] = t
    return out


def convert_deep_supervision_dict_to_list(tensor_dict: dict[str, torch.Tensor]) -> list[torch.Tensor]:
    """Inverse of the above: full-resolution head first, then ds heads in
    index order."""
    keys = sorted(tensor_dict.keys(), key=lambda k: (k != "prediction", k))
    return [tensor_dict[k] for k in keys]


def prepare_loss_arg(
    arg: torch.Tensor | dict[str, torch.Tensor],
) -> torch.Tensor | list[torch.Tensor]:
    """Loss-call argument normalisation (reference :283-305): a single tensor
    passes through; a dict with one entry unwraps; a deep-supervision dict
    becomes the ordered list the pyramid loss expects."""
    if isinstance(arg, torch.Tensor):
        return arg
    if len(arg) == 1:
        return next(iter(arg.values()))
    return convert_deep_supervision_dict_to_list(arg)


def get_segs_from_probs(preds: torch.Tensor, has_regions: bool = False, threshold: float = 0.5) -> torch.Tensor:
    """Probabilities [B, C, ...] -> one-hot segmentation (reference :213-245).
    With region-based labels each class is thresholded independently; else
    argmax one-hot."""
    if has_regions:
        return preds > threshold
    argmax = preds.argmax(dim=1, keepdim=True)
    seg = torch.zeros_like(preds, dtype=torch.bool)
    seg.scatter_(1, argmax, True)
    return seg


def collapse_one_hot_tensor(input: torch.Tensor, dim: int = 0) -> torch.Tensor:
    """One-hot -> integer labels along dim (reference :247-259)."""
    return torch.argmax(input.long(), dim=dim)


def get_dataset_n_voxels(source_plans: dict, n_cases: int) -> float:
    """Total voxels in the dataset from the plans' median shape (reference
    :261-281; used for the 5% batch-size cap)."""
    cfgs = source_plans["configurations"]
    key = "3d_fullres" if "3d_fullres" in cfgs else "2d"
    import numpy as np

    return float(np.prod(cfgs[key]["median_image_size_in_voxels"], dtype=np.float64)) * n_cases


# ---------------------------------------------------------------------------
# async multiprocess patch loader (reference NnUNetDataLoaderWrapper :307-438:
# wraps nnunet's MultiThreadedAugmenter; ours wraps either that or the native
# _PatchWorker process — both with EXPLICIT shutdown of child processes)
# ---------------------------------------------------------------------------


def _patch_worker_main(queue: mp.Queue, stop, volumes, segs, patch_size, batch_size, ds_levels, seed):
    """Child process: sample foreground-oversampled patches forever."""
    from fl4health_amd.preprocessing.nnunet import downsample_seg_pyramid, sample_patch

    gen = torch.Generator().manual_seed(seed)
    try:
        while not stop.is_set():
            idxs = [int(torch.randint(0, len(volumes), (), generator=gen)) for _ in range(batch_size)]
            xs, ys = [], []
            for i in idxs:
                x, y = sample_patch(volumes[i], segs[i], patch_size, True, gen)
                xs.append(x)
                ys.append(y)
            xb = torch.stack(xs)
            yb = torch.stack(ys)
            if ds_levels > 1:
                targets = downsample_seg_pyramid(yb, ds_levels)
            else:
                targets = yb
            try:
                queue.put((xb, targets), timeout=1.0)
            except Exception:  # noqa: BLE001 — queue full: check stop and retry
                continue
    except KeyboardInterrupt:
        pass


class AsyncPatchLoader:
    """Background-process patch sampler for 3D volumes: the augmentation /
    sampling pipeline runs in a separate process feeding a bounded queue
    (the MI355X-native analogue of nnunet's MultiThreadedAugmenter), with an
    explicit ``shutdown`` that terminates the child (reference
    clients/nnunet_client.py:883-913 lifecycle requirement)."""

    def __init__(
        self,
        volumes: list[torch.Tensor],
        segs: list[torch.Tensor],
        patch_size: Sequence[int],
        batch_size: int,
        n_batches_per_epoch: int = 50,
        deep_supervision_levels: int = 1,
        seed: int = 0,
        queue_depth: int = 4,
    ) -> None:
        self.batch_size = batch_size
        self.n_batches = n_batches_per_epoch
        ctx = mp.get_context("spawn")
        self._queue: mp.Queue = ctx.Queue(maxsize=queue_depth)
        self._stop = ctx.Event()
        self._proc = ctx.Process(
            target=_patch_worker_main,
            args=(self._queue, self._stop, volumes, segs, list(patch_size), batch_size,
                  deep_supervision_levels, seed),
            daemon=True,
        )
        self._proc.start()
        self._alive = True

    def __len__(self) -> int:
        return self.n_batches

    def __iter__(self):
        assert self._alive, "loader has been shut down"
        for _ in range(self.n_batches):
            yield self._queue.get()

    def shutdown(self) -> None:
        if not self._alive:
            return
        self._alive = False
        self._stop.set()
        try:
            while not self._queue.empty():
                self._queue.get_nowait()
        except Exception:  # noqa: BLE001
            pass
        self._proc.join(timeout=5.0)
        if self._proc.is_alive():
            self._proc.terminate()
            self._proc.join(timeout=5.0)
        self._queue.close()

    def __del__(self) -> None:  # last-resort cleanup
        try:
            self.shutdown()
        except Exception:  # noqa: BLE001
            pass


class NnUNetDataLoaderWrapper(DataLoader):
    """Adapts a generator-style nnunet/native loader to the
    ``torch.utils.data.DataLoader`` interface BasicClient consumes, including
    deep-supervision target dicts and graceful multiprocess shutdown
    (reference nnunet_utils.py:307-438)."""

    def __init__(self, nnunet_augmenter: Any, nnunet_config: NnunetConfig | str,
                 infinite: bool = False) -> None:
        self.nnunet_augmenter = nnunet_augmenter
        config = NnunetConfig(nnunet_config) if isinstance(nnunet_config, str) else nnunet_config
        self.num_spatial_dims = NNUNET_N_SPATIAL_DIMS[config]
        self.infinite = infinite
        self.current_step = 0
        # mirror DataLoader surface without invoking its worker machinery
        self._len = len(nnunet_augmenter) if hasattr(nnunet_augmenter, "__len__") else 250
        bs = int(getattr(nnunet_augmenter, "batch_size", 1))
        self.dataset = getattr(nnunet_augmenter, "data_loader", None) or range(self._len * bs)

    def __len__(self) -> int:
        return self._len

    def _convert(self, batch: Any) -> tuple[torch.Tensor, torch.Tensor | dict[str, torch.Tensor]]:
        if isinstance(batch, dict):  # nnunet augmenter batches
            inputs, targets = batch["data"], batch["target"]
        else:
            inputs, targets = batch
        if isinstance(targets, (list, tuple)):
            target_dict = convert_deep_supervision_list_to_dict(list(targets), self.num_spatial_dims)
            return inputs, target_dict
        return inputs, targets

    def __iter__(self):
        self.current_step = 0
        if hasattr(self.nnunet_augmenter, "__iter__"):
            for batch in self.nnunet_augmenter:
                yield self._convert(batch)
        else:  # generator protocol: next() forever
            while self.infinite or self.current_step < self._len:
                self.current_step += 1
                yield self._convert(next(self.nnunet_augmenter))

    def reset(self) -> None:
        self.current_step = 0

    def shutdown(self) -> None:
        """Terminate the augmenter's child processes (reference :429-438)."""
        aug = self.nnunet_augmenter
        if hasattr(aug, "shutdown"):
            aug.shutdown()
        elif hasattr(aug, "_finish"):  # nnunet MultiThreadedAugmenter
            aug._finish()
        del self.nnunet_augmenter


class Module2LossWrapper(_Loss):
    """nn.Module loss -> _Loss (reference :440-465)."""

    def __init__(self, loss: nn.Module, **kwargs: Any) -> None:
        super().__init__(**kwargs)
        self.loss = loss

    def forward(self, pred: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
        return self.loss(pred, target)


class StreamToLogger(io.StringIO):
    """Redirect a stdout-ish stream into a logger (reference :467-489);
    used to keep nnunet's prints out of the FL logs unless debugging."""

    def __init__(self, logger: Logger, level: int) -> None:
        super().__init__()
        self.logger = logger
        self.level = level
        self.linebuf = ""

    def write(self, buf: str) -> int:
        for line in buf.rstrip().splitlines():
            self.logger.log(self.level, line.rstrip())
        return len(buf)

    def flush(self) -> None:
        pass


class PolyLRSchedulerWrapper(_LRScheduler):
    """Polynomial decay by STEP with the current-torch signature (reference
    :491-546; nnunet's own scheduler predates the signature change)."""

    def __init__(self, optimizer: torch.optim.Optimizer, initial_lr: float, max_steps: int,
                 exponent: float = 0.9) -> None:
        self.initial_lr = initial_lr
        self.max_steps = max_steps
        self.exponent = exponent
        self._step_count_local = 0
        super().__init__(optimizer)

    def get_lr(self) -> list[float]:
        step = min(self._step_count - 1, self.max_steps - 1)
        return [self.initial_lr * (1 - step / self.max_steps) ** self.exponent for _ in self.optimizer.param_groups]
