"""Data layers: DummyData, MemoryData, Data (PDB/LMDB), ImageData,
HDF5Data/HDF5Output, WindowData.

Parity: /root/reference/src/caffe/layers/{dummy_data,memory_data,data,
image_data,hdf5_data,hdf5_output,window_data}_layer.cpp and
base_data_layer.cpp (background prefetch thread + double buffer).

Distributed sharding keeps the reference's striped semantics (SURVEY P4,
data_layer.cpp:143-161): rank r of W ranks reads records r, r+W, r+2W, ...
(here one process == one GPU, so thread striping collapses to rank striping).
"""

from __future__ import annotations

import os
import threading
from typing import List, Optional

import numpy as np
import torch

from ..core.context import ctx
from ..core.layer import Layer, register_layer
from ..core import filler as fillers
from ..data.pdb import PDBReader, datum_to_array
from ..data.transformer import DataTransformer
from ..proto import Message


@register_layer("DUMMY_DATA")
class DummyDataLayer(Layer):
    exact_num_bottom = 0

    def layer_setup(self, bottom, top) -> None:
        dp = self.param.ensure("dummy_data_param")
        n_top = len(top)
        fparams = list(dp.data_filler)
        if len(fparams) == 0:
            fparams = [Message("FillerParameter", type="constant", value=0.0)]
        if len(fparams) == 1:
            fparams = fparams * n_top
        self.fillers = fparams
        # refill each forward only for non-constant fillers (dummy_data_layer.cpp)
        self.refill = [fp.type != "constant" for fp in fparams]

        def dim(lst, i):
            vals = list(lst)
            if not vals:
                return None
            return int(vals[i]) if len(vals) > 1 else int(vals[0])

        self.shapes = []
        for i in range(n_top):
            shape = tuple(d for d in (dim(dp.num, i), dim(dp.channels, i),
                                      dim(dp.height, i), dim(dp.width, i))
                          if d is not None)
            self.shapes.append(shape)
        self._filled = False

    def reshape(self, bottom, top) -> None:
        for t, s in zip(top, self.shapes):
            t.reshape(s)

    def forward(self, bottom, top) -> None:
        cd = ctx().compute_dtype
        for i, t in enumerate(top):
            if not self._filled or self.refill[i]:
                fillers.fill(t, self.fillers[i])
                # data top (first) runs at the compute dtype; label-like tops
                # stay fp32 (class ids > 256 are inexact in bf16)
                if i == 0 and t.data.dtype != cd:
                    t.data = t.data.to(cd)
        self._filled = True

    def backward(self, top, propagate_down, bottom) -> None:
        pass


@register_layer("MEMORY_DATA")
class MemoryDataLayer(Layer):
    exact_num_bottom = 0
    exact_num_top = 2

    def layer_setup(self, bottom, top) -> None:
        mp = self.param.ensure("memory_data_param")
        self.batch = int(mp.batch_size)
        self.c, self.h, self.w = int(mp.channels), int(mp.height), int(mp.width)
        self._data: Optional[torch.Tensor] = None
        self._labels: Optional[torch.Tensor] = None
        self._pos = 0

    def add_data(self, data: torch.Tensor, labels: torch.Tensor) -> None:
        assert data.shape[0] % self.batch == 0
        self._data = data
        self._labels = labels
        self._pos = 0

    # alias matching the reference's MemoryDataLayer::Reset
    reset = add_data

    def reshape(self, bottom, top) -> None:
        top[0].reshape(self.batch, self.c, self.h, self.w)
        top[1].reshape(self.batch)

    def forward(self, bottom, top) -> None:
        if self._data is None:
            raise RuntimeError("MemoryDataLayer: call add_data() first")
        n = self._data.shape[0]
        idx = torch.arange(self._pos, self._pos + self.batch) % n
        c = ctx()
        top[0].data = self._data[idx].to(c.torch_device, c.compute_dtype)
        top[1].data = self._labels[idx].to(c.torch_device, torch.float32)
        self._pos = (self._pos + self.batch) % n

    def backward(self, top, propagate_down, bottom) -> None:
        pass


class _PrefetchingDataLayer(Layer):
    """Base for DB-backed layers: a host thread assembles the next batch
    (decode + transform) while the GPU trains on the current one
    (base_data_layer.cpp:56-105)."""

    exact_num_bottom = 0

    def _start_prefetch(self) -> None:
        self._ready = threading.Event()
        self._taken = threading.Event()
        self._taken.set()
        self._stop = False
        self._batch = None
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()

    def _loop(self) -> None:
        while not self._stop:
            self._taken.wait()
            self._taken.clear()
            self._batch = self._load_batch()
            self._ready.set()

    def _next_batch(self):
        self._ready.wait()
        self._ready.clear()
        batch = self._batch
        self._taken.set()
        return batch

    def _load_batch(self):
        raise NotImplementedError

    def backward(self, top, propagate_down, bottom) -> None:
        pass


@register_layer("DATA")
class DataLayer(_PrefetchingDataLayer):
    """Reads Datum records from a PDB file (or LMDB if the lib is present).

    Striped sharding: with shared_file_system (one DB for all ranks) rank r
    starts at record r and strides by world_size; without it, each rank opens
    source_<r> (per-client shards, the reference's source_k convention,
    data_layer.cpp:232-261)."""

    def layer_setup(self, bottom, top) -> None:
        dp = self.param.ensure("data_param")
        self.batch = int(dp.batch_size)
        c = ctx()
        source = dp.source
        backend = dp.enum_name("backend") if dp.has("backend") else "LEVELDB"
        shared = bool(dp.shared_file_system) or c.world_size == 1
        if not shared:
            source = f"{source}_{c.rank}"
        # Reference-produced environments are read directly by the
        # pure-python walkers: LMDB (data/lmdb_io.py) and LevelDB
        # incl. snappy blocks + log replay (data/leveldb_io.py); PDB
        # files keep the native container path.
        is_lmdb = (os.path.isdir(source)
                   and os.path.exists(os.path.join(source, "data.mdb"))) or (
                  os.path.isfile(source) and source.endswith(".mdb"))
        if is_lmdb:
            from ..data.lmdb_io import LmdbReader
            self.db = LmdbReader(source)
        elif os.path.isdir(source):
            from ..data.leveldb_io import LevelDbReader
            self.db = LevelDbReader(source)
        else:
            self.db = PDBReader(source)
        self.stride = c.world_size if shared else 1
        self.cursor = c.rank if shared else 0
        rng = np.random.default_rng(c.seed + 131 * c.rank)
        self.transform = DataTransformer(self.param.transform_param,
                                         self.phase, rng)
        first = datum_to_array(self.db.get(0))
        sample = self.transform(first)
        self.shape = sample.shape
        self._start_prefetch()

    def _load_batch(self):
        data = np.empty((self.batch,) + self.shape, dtype=np.float32)
        labels = np.empty((self.batch,), dtype=np.float32)
        for i in range(self.batch):
            d = self.db.get(self.cursor % len(self.db))
            data[i] = self.transform(datum_to_array(d))
            labels[i] = d.label or 0
            self.cursor += self.stride
        return data, labels

    def reshape(self, bottom, top) -> None:
        top[0].reshape((self.batch,) + self.shape)
        if len(top) > 1:
            top[1].reshape(self.batch)

    def forward(self, bottom, top) -> None:
        data, labels = self._next_batch()
        c = ctx()
        top[0].data = torch.from_numpy(data).to(c.torch_device, c.compute_dtype,
                                                non_blocking=True)
        if len(top) > 1:
            top[1].data = torch.from_numpy(labels).to(c.torch_device,
                                                      non_blocking=True)


@register_layer("IMAGE_DATA")
class ImageDataLayer(_PrefetchingDataLayer):
    """File-list image reader (image_data_layer.cpp). Uses PIL for decode;
    rank-striped partitioning of the list (thread_global_idx semantics)."""

    def layer_setup(self, bottom, top) -> None:
        ip = self.param.ensure("image_data_param")
        self.batch = int(ip.batch_size)
        self.new_h, self.new_w = int(ip.new_height), int(ip.new_width)
        c = ctx()
        source = ip.source
        shared = bool(ip.shared_file_system) or c.world_size == 1
        if not shared:
            source = f"{source}_{c.rank}"
        with open(source) as f:
            lines = [l.split() for l in f if l.strip()]
        self.entries = [(p, int(lbl)) for p, lbl in lines]
        if shared and c.world_size > 1:
            self.entries = self.entries[c.rank::c.world_size]
        if bool(ip.shuffle):
            rng = np.random.default_rng(c.seed + c.rank)
            rng.shuffle(self.entries)
        self.cursor = 0
        rng = np.random.default_rng(c.seed + 977 * c.rank)
        self.transform = DataTransformer(self.param.transform_param,
                                         self.phase, rng)
        sample = self.transform(self._read(self.entries[0][0]))
        self.shape = sample.shape
        self._start_prefetch()

    def _read(self, path: str) -> np.ndarray:
        from PIL import Image
        img = Image.open(path).convert("RGB")
        if self.new_h and self.new_w:
            img = img.resize((self.new_w, self.new_h))
        arr = np.asarray(img, dtype=np.float32)  # HWC RGB
        # Caffe stores BGR; match channel order for mean-file compat
        return np.ascontiguousarray(arr[:, :, ::-1].transpose(2, 0, 1))

    def _load_batch(self):
        data = np.empty((self.batch,) + self.shape, dtype=np.float32)
        labels = np.empty((self.batch,), dtype=np.float32)
        for i in range(self.batch):
            path, lbl = self.entries[self.cursor % len(self.entries)]
            data[i] = self.transform(self._read(path))
            labels[i] = lbl
            self.cursor += 1
        return data, labels

    def reshape(self, bottom, top) -> None:
        top[0].reshape((self.batch,) + self.shape)
        top[1].reshape(self.batch)

    def forward(self, bottom, top) -> None:
        data, labels = self._next_batch()
        c = ctx()
        top[0].data = torch.from_numpy(data).to(c.torch_device, c.compute_dtype,
                                                non_blocking=True)
        top[1].data = torch.from_numpy(labels).to(c.torch_device,
                                                  non_blocking=True)


@register_layer("HDF5_DATA")
class HDF5DataLayer(Layer):
    """Reads "data"/"label" datasets from the HDF5 files listed in the
    source file, cycling through files and rows
    (hdf5_data_layer.cpp:27-112; contiguous-layout HDF5 parsed by the
    pure-python data/hdf5_io.py reader -- no libhdf5 in this image)."""

    exact_num_bottom = 0

    def layer_setup(self, bottom, top) -> None:
        from ..data.hdf5_io import Hdf5Reader
        hp = self.param.ensure("hdf5_data_param")
        self.batch = int(hp.batch_size)
        with open(hp.source) as f:
            self.files = [ln.strip() for ln in f if ln.strip()]
        if not self.files:
            raise ValueError(f"{hp.source}: empty HDF5 source list")
        self._file_idx = 0
        self._row = 0
        self._load(Hdf5Reader(self.files[0]))

    def _load(self, r) -> None:
        self._data = r.get("data").astype(np.float32)
        self._label = r.get("label").astype(np.float32).reshape(-1)
        if self._data.shape[0] != self._label.shape[0]:
            raise ValueError("data/label row mismatch")

    def reshape(self, bottom, top) -> None:
        shape = (self.batch,) + tuple(self._data.shape[1:])
        top[0].reshape(shape if len(shape) == 4 else
                       shape + (1,) * (4 - len(shape)))
        if len(top) > 1:
            top[1].reshape(self.batch)

    def forward(self, bottom, top) -> None:
        from ..data.hdf5_io import Hdf5Reader
        rows = []
        labels = []
        for _ in range(self.batch):
            if self._row >= self._data.shape[0]:
                self._file_idx = (self._file_idx + 1) % len(self.files)
                self._load(Hdf5Reader(self.files[self._file_idx]))
                self._row = 0
            rows.append(self._data[self._row])
            labels.append(self._label[self._row])
            self._row += 1
        c = ctx()
        batch = np.stack(rows)
        top[0].data = torch.from_numpy(batch).to(
            c.torch_device, c.compute_dtype).view(top[0].shape)
        if len(top) > 1:
            top[1].data = torch.tensor(labels, dtype=torch.float32,
                                       device=c.torch_device)

    def backward(self, top, propagate_down, bottom) -> None:
        pass


@register_layer("HDF5_OUTPUT")
class HDF5OutputLayer(Layer):
    """Accumulates (data, label) bottoms across forwards and writes them
    as the "data"/"label" datasets of file_name on finalize()
    (hdf5_output_layer.cpp:16-44)."""

    exact_num_bottom = 2
    exact_num_top = 0

    def layer_setup(self, bottom, top) -> None:
        op = self.param.ensure("hdf5_output_param")
        self.file_name = op.file_name
        self._rows: list = []
        self._labels: list = []

    def reshape(self, bottom, top) -> None:
        pass

    def forward(self, bottom, top) -> None:
        self._rows.append(
            bottom[0].data.detach().to(torch.float32).cpu().numpy().copy())
        self._labels.append(
            bottom[1].data.detach().to(torch.float32).cpu().numpy().copy())

    def finalize(self) -> None:
        from ..data.hdf5_io import Hdf5Writer
        with Hdf5Writer(self.file_name) as w:
            w.put("data", np.concatenate(self._rows))
            w.put("label", np.concatenate(self._labels))

    def backward(self, top, propagate_down, bottom) -> None:
        pass


@register_layer("WINDOW_DATA")
class WindowDataLayer(_PrefetchingDataLayer):
    """R-CNN-style window sampling (window_data_layer.cpp): a window file
    lists images and class/overlap/box windows; each batch draws
    fg_fraction foreground windows (overlap >= fg_threshold) and fills the
    rest with background (overlap < bg_threshold, class 0), crops with
    context_pad and warps to crop_size.

    Window-file format (the reference's):
        # <image_index>
        <image_path>
        <channels> <height> <width>
        <num_windows>
        <class> <overlap> <x1> <y1> <x2> <y2>
    """

    def layer_setup(self, bottom, top) -> None:
        wp = self.param.ensure("window_data_param")
        self.batch = int(wp.batch_size)
        self.crop = int(wp.crop_size)
        assert self.crop > 0, "WINDOW_DATA needs crop_size"
        self.fg_thresh = float(wp.fg_threshold)
        self.bg_thresh = float(wp.bg_threshold)
        self.fg_frac = float(wp.fg_fraction)
        self.ctx_pad = int(wp.context_pad)
        self.scale = float(wp.scale)
        self.mirror = bool(wp.mirror)
        self.mean = None
        if wp.has("mean_file"):
            from ..proto import read_proto_binary
            proto = read_proto_binary(wp.mean_file, "BlobProto")
            arr = np.asarray(proto.data, dtype=np.float32)
            self.mean = arr.reshape(proto.channels, proto.height, proto.width)

        self.images: List[str] = []
        self.fg: List[tuple] = []   # (img_idx, cls, x1, y1, x2, y2)
        self.bg: List[tuple] = []
        with open(wp.source) as f:
            lines = [l.rstrip("\n") for l in f]
        i = 0
        while i < len(lines):
            if not lines[i].startswith("#"):
                i += 1
                continue
            path = lines[i + 1].strip()
            nwin = int(lines[i + 3])
            img_idx = len(self.images)
            self.images.append(path)
            for w in range(nwin):
                parts = lines[i + 4 + w].split()
                cls, ov = int(parts[0]), float(parts[1])
                box = tuple(int(v) for v in parts[2:6])
                if ov >= self.fg_thresh:
                    self.fg.append((img_idx, cls) + box)
                elif ov < self.bg_thresh:
                    self.bg.append((img_idx, 0) + box)
            i += 4 + nwin
        if not self.fg or not self.bg:
            raise ValueError("window file needs both fg and bg windows")
        c = ctx()
        self.rng = np.random.default_rng(c.seed + 389 * c.rank)
        self._img_cache: dict = {}
        self._start_prefetch()

    def _read_img(self, idx):
        if idx not in self._img_cache:
            from PIL import Image
            img = Image.open(self.images[idx]).convert("RGB")
            arr = np.asarray(img, dtype=np.float32)[:, :, ::-1]  # BGR
            self._img_cache[idx] = np.ascontiguousarray(arr)
            if len(self._img_cache) > 64:
                self._img_cache.pop(next(iter(self._img_cache)))
        return self._img_cache[idx]

    def _crop_window(self, win):
        from PIL import Image
        img_idx, cls, x1, y1, x2, y2 = win
        arr = self._read_img(img_idx)
        H, W = arr.shape[:2]
        if self.ctx_pad:
            pw = int(round((x2 - x1 + 1) * self.ctx_pad / self.crop))
            ph = int(round((y2 - y1 + 1) * self.ctx_pad / self.crop))
            x1, x2 = x1 - pw, x2 + pw
            y1, y2 = y1 - ph, y2 + ph
        x1c, y1c = max(0, x1), max(0, y1)
        x2c, y2c = min(W - 1, x2), min(H - 1, y2)
        patch = arr[y1c:y2c + 1, x1c:x2c + 1]
        im = Image.fromarray(patch.astype(np.uint8))
        im = im.resize((self.crop, self.crop))  # warp mode
        out = np.asarray(im, dtype=np.float32).transpose(2, 0, 1)
        if self.mean is not None:
            out = out - self.mean[:, :self.crop, :self.crop]
        if self.mirror and self.rng.integers(0, 2):
            out = out[:, :, ::-1]
        return np.ascontiguousarray(out * self.scale), cls

    def _load_batch(self):
        n_fg = int(round(self.batch * self.fg_frac))
        data = np.empty((self.batch, 3, self.crop, self.crop), dtype=np.float32)
        labels = np.empty((self.batch,), dtype=np.float32)
        for i in range(self.batch):
            pool = self.fg if i < n_fg else self.bg
            win = pool[int(self.rng.integers(0, len(pool)))]
            data[i], labels[i] = self._crop_window(win)
        return data, labels

    def reshape(self, bottom, top) -> None:
        top[0].reshape(self.batch, 3, self.crop, self.crop)
        top[1].reshape(self.batch)

    def forward(self, bottom, top) -> None:
        data, labels = self._next_batch()
        c = ctx()
        top[0].data = torch.from_numpy(data).to(c.torch_device, c.compute_dtype)
        top[1].data = torch.from_numpy(labels).to(c.torch_device)
