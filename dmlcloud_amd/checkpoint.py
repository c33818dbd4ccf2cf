"""Checkpoint directories and fast tensor state serialization.

Directory layout parity with the reference (reference
dmlcloud/checkpoint.py:12-123): `{root}/{name}-{YYYY.MM.DD-HH.MM}-{id}/`
containing `.dmlcloud` indicator, `config.yaml`, `log.txt`, optional
`.slurm-jobid`; SLURM-requeue rediscovery via the job-id file.

Beyond the reference (which accepts save_latest/save_interval/save_best
on register_model but never implements weight saving — reference
pipeline.py:61-64), this module implements the declared
checkpointing with an MI355X-native fast path (`.dmlt` format):

  save: every device tensor in the (arbitrary, nested) state object is
        packed into ONE flat HBM buffer by the gfx950 chunked-copy kernel
        (ops/csrc/copy.hip), followed by a single pinned D2H copy and one
        sequential file write — instead of one D2H + write per tensor.
  load: one file read + one H2D upload; tensors are reconstructed as
        zero-copy views of the flat device buffer.
"""

import datetime
import io
import logging
import pickle
import secrets
from pathlib import Path
from typing import Optional, Union

import torch

from . import ops
from .config import Config
from .utils.slurm import slurm_job_id

_MAGIC = b'DMLT0001'
_ALIGN = 16


def sanitize_filename(filename: str) -> str:
    return filename.replace('/', '_')


def generate_id() -> str:
    s = secrets.token_urlsafe(5)
    return s.replace('-', 'a').replace('_', 'b')


def generate_checkpoint_path(
    root: Union[Path, str], name: Optional[str] = None, creation_time: Optional[datetime.datetime] = None
) -> Path:
    root = Path(root)
    if name is None:
        name = 'run'
    if creation_time is None:
        creation_time = datetime.datetime.now()
    dt = creation_time.strftime('%Y.%m.%d-%H.%M')
    return root / f'{sanitize_filename(name)}-{dt}-{generate_id()}'


def find_slurm_checkpoint(root: Union[Path, str]) -> Optional[Path]:
    """Find a checkpoint dir created by this SLURM job (requeue recovery)."""
    root = Path(root)
    job_id = slurm_job_id()
    if job_id is None or not root.exists():
        return None
    for child in root.iterdir():
        ckpt = CheckpointDir(child)
        if ckpt.is_valid and ckpt.slurm_job_id == job_id:
            return child
    return None


class CheckpointDir:
    def __init__(self, path: Union[Path, str]):
        self.path = Path(path).resolve()
        self.logger = logging.getLogger('dmlcloud_amd')

    @property
    def config_file(self) -> Path:
        return self.path / 'config.yaml'

    @property
    def indicator_file(self) -> Path:
        return self.path / '.dmlcloud'

    @property
    def log_file(self) -> Path:
        return self.path / 'log.txt'

    @property
    def slurm_file(self) -> Path:
        return self.path / '.slurm-jobid'

    @property
    def models_dir(self) -> Path:
        return self.path / 'models'

    @property
    def state_file(self) -> Path:
        return self.path / 'state.dmlt'

    @property
    def exists(self) -> bool:
        return self.path.exists()

    @property
    def is_valid(self) -> bool:
        return self.exists and self.path.is_dir() and self.indicator_file.exists()

    @property
    def slurm_job_id(self) -> Optional[str]:
        if not self.slurm_file.exists():
            return None
        return self.slurm_file.read_text()

    def create(self):
        if self.exists:
            raise ValueError(f'Checkpoint directory already exists: {self.path}')
        self.path.mkdir(parents=True, exist_ok=True)
        self.indicator_file.touch()
        self.log_file.touch()
        if slurm_job_id() is not None:
            self.slurm_file.write_text(slurm_job_id())

    def save_config(self, config: Config):
        if not self.exists:
            raise ValueError(f'Checkpoint directory does not exist: {self.path}')
        Config.create(config).save(self.config_file)

    def load_config(self) -> Config:
        if not self.is_valid:
            raise ValueError(f'Checkpoint directory is not valid: {self.path}')
        return Config.load(self.config_file)

    def __str__(self):
        return str(self.path)

    def __repr__(self):
        return f'CheckpointDir({self.path})'


# --------------------------------------------------------------------------
# .dmlt tensor-state format
# --------------------------------------------------------------------------


class _TensorRef:
    __slots__ = ('index',)

    def __init__(self, index: int):
        self.index = index

    def __reduce__(self):
        return (_TensorRef, (self.index,))


def _extract_tensors(obj, tensors: list):
    """Replace every tensor in a nested container with a _TensorRef."""
    if isinstance(obj, torch.Tensor):
        tensors.append(obj)
        return _TensorRef(len(tensors) - 1)
    if isinstance(obj, dict):
        return {k: _extract_tensors(v, tensors) for k, v in obj.items()}
    if isinstance(obj, (list, tuple)):
        mapped = [_extract_tensors(v, tensors) for v in obj]
        return mapped if isinstance(obj, list) else tuple(mapped)
    return obj


def _insert_tensors(obj, tensors: list):
    if isinstance(obj, _TensorRef):
        return tensors[obj.index]
    if isinstance(obj, dict):
        return {k: _insert_tensors(v, tensors) for k, v in obj.items()}
    if isinstance(obj, (list, tuple)):
        mapped = [_insert_tensors(v, tensors) for v in obj]
        return mapped if isinstance(obj, list) else tuple(mapped)
    return obj


def _align(n: int) -> int:
    return (n + _ALIGN - 1) // _ALIGN * _ALIGN


_DTYPE_NAMES = {str(dt).replace('torch.', ''): dt for dt in (
    torch.float32, torch.float64, torch.float16, torch.bfloat16, torch.int8, torch.uint8,
    torch.int16, torch.int32, torch.int64, torch.bool,
)}


def _byte_view(t: torch.Tensor) -> torch.Tensor:
    t = t.detach().contiguous().reshape(-1)  # 0-dim tensors cannot be dtype-viewed
    if t.dtype is torch.bool:
        t = t.view(torch.uint8)
    return t.view(torch.uint8) if t.dtype is not torch.uint8 else t


def save_tensor_state(obj, path: Union[str, Path]):
    """Serialize a nested state object (dicts/lists/tensors/scalars) to
    `path` in the .dmlt format with a single fused device pack."""
    tensors = []
    skeleton = _extract_tensors(obj, tensors)

    metas = []
    offset = 0
    for t in tensors:
        nbytes = t.numel() * t.element_size()
        metas.append(
            {
                'shape': list(t.shape),
                'dtype': str(t.dtype).replace('torch.', ''),
                'offset': offset,
                'nbytes': nbytes,
            }
        )
        offset = _align(offset + nbytes)
    total = offset

    host_buf = torch.empty(total, dtype=torch.uint8, pin_memory=torch.cuda.is_available()) if total else None

    # zero-element tensors carry no payload (their meta reconstructs them)
    device_tensors = [(i, t) for i, t in enumerate(tensors) if t.is_cuda and t.numel()]
    cpu_tensors = [(i, t) for i, t in enumerate(tensors) if not t.is_cuda and t.numel()]

    if device_tensors:
        dev = device_tensors[0][1].device
        flat_dev = torch.empty(total, dtype=torch.uint8, device=dev)
        srcs, dsts = [], []
        for i, t in device_tensors:
            bv = _byte_view(t)
            srcs.append(bv)
            dsts.append(flat_dev[metas[i]['offset'] : metas[i]['offset'] + metas[i]['nbytes']])
        ops.chunked_copy(srcs, dsts)  # one gfx950 kernel packs everything
        host_buf.copy_(flat_dev, non_blocking=True)
        torch.cuda.synchronize(dev)

    for i, t in cpu_tensors:
        bv = _byte_view(t)
        host_buf[metas[i]['offset'] : metas[i]['offset'] + metas[i]['nbytes']].copy_(bv.reshape(-1))

    header = pickle.dumps({'skeleton': skeleton, 'metas': metas, 'total': total})
    path = Path(path)
    path.parent.mkdir(parents=True, exist_ok=True)
    tmp = path.with_suffix(path.suffix + '.tmp')
    with open(tmp, 'wb') as f:
        f.write(_MAGIC)
        f.write(len(header).to_bytes(8, 'little'))
        f.write(header)
        if total:
            # memoryview over the pinned host buffer: no duplicate of the
            # (potentially multi-GB) payload before the write syscall
            f.write(memoryview(host_buf.numpy()))
    tmp.replace(path)  # atomic publish


def load_tensor_state(path: Union[str, Path], device: Union[str, torch.device, None] = None):
    """Load a .dmlt file. With a CUDA device, the payload is uploaded with
    ONE H2D copy and tensors are zero-copy views of the flat buffer."""
    import numpy as np

    path = Path(path)
    with open(path, 'rb') as f:
        magic = f.read(8)
        if magic != _MAGIC:
            raise ValueError(f'{path} is not a .dmlt checkpoint (bad magic {magic!r})')
        header_len = int.from_bytes(f.read(8), 'little')
        header = pickle.loads(f.read(header_len))
        total = header['total']
        # read straight into the destination buffer: one file read, no
        # intermediate bytes object for the payload
        host_np = np.empty(total, dtype=np.uint8)
        got = f.readinto(memoryview(host_np))
        if got != total:
            raise ValueError(f'{path}: truncated payload ({got} of {total} bytes)')

    metas = header['metas']
    host_flat = torch.from_numpy(host_np) if total else torch.empty(0, dtype=torch.uint8)

    device = torch.device(device) if device is not None else None
    if device is not None and device.type == 'cuda':
        flat = host_flat.to(device, non_blocking=False)
    else:
        flat = host_flat

    tensors = []
    for meta in metas:
        dtype = _DTYPE_NAMES[meta['dtype']]
        raw = flat[meta['offset'] : meta['offset'] + meta['nbytes']]
        if dtype is torch.bool:
            t = raw.view(torch.uint8).reshape(meta['shape']).to(torch.bool)
        else:
            t = raw.view(dtype).reshape(meta['shape'])
        tensors.append(t)

    return _insert_tensors(header['skeleton'], tensors)


class ModelCheckpointer:
    """Implements the save_latest / save_interval / save_best policy that
    the reference's register_model promises (reference pipeline.py:61-64)
    but never executes."""

    def __init__(
        self,
        checkpoint_dir: CheckpointDir,
        name: str,
        save_latest: bool = True,
        save_interval: Optional[int] = None,
        save_best: bool = False,
        best_metric: str = 'val/loss',
        higher_is_better: bool = False,
    ):
        self.checkpoint_dir = checkpoint_dir
        self.name = name
        self.save_latest = save_latest
        self.save_interval = save_interval
        self.save_best = save_best
        self.best_metric = best_metric
        self.higher_is_better = higher_is_better
        self.best_value = None

    def _model_dir(self) -> Path:
        return self.checkpoint_dir.models_dir / sanitize_filename(self.name)

    def maybe_save(self, model: torch.nn.Module, epoch: int, tracker=None):
        """Root-only policy evaluation + save. `model` may be DDP-wrapped."""
        module = model.module if hasattr(model, 'module') else model
        state = {'epoch': epoch, 'state_dict': module.state_dict()}
        directory = self._model_dir()

        if self.save_latest:
            save_tensor_state(state, directory / 'latest.dmlt')
        if self.save_interval and epoch % self.save_interval == 0:
            save_tensor_state(state, directory / f'epoch_{epoch:04d}.dmlt')
        if self.save_best and tracker is not None and self.best_metric in tracker:
            history = tracker[self.best_metric]
            if history and history[-1] is not None:
                value = float(history[-1])
                better = (
                    self.best_value is None
                    or (value > self.best_value if self.higher_is_better else value < self.best_value)
                )
                if better:
                    self.best_value = value
                    save_tensor_state(state, directory / 'best.dmlt')

    def load(self, which: str = 'latest', device=None):
        return load_tensor_state(self._model_dir() / f'{which}.dmlt', device=device)


def _pickle_safe(obj):
    """Round-trip check helper (used by tests)."""
    buf = io.BytesIO()
    pickle.dump(obj, buf)
    buf.seek(0)
    return pickle.load(buf)
