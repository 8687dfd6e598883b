"""Python API of the MI355X-native shared-tensor engine.

Mirrors the reference's three-call surface (createOrFetch / copyToTensor /
addFromTensor, /root/reference/src/sharedtensor.c:347-453) on top of the
native engine (csrc/engine.cpp), and extends it with the features the
reference's README names as missing (README.md:29-47): table-of-tensors sync
with per-tensor scales, real GPU codec kernels, bandwidth limiting,
reconnection, observability.

All large buffers (replica, per-link residual deltas, message staging) are
torch tensors allocated here and handed to the engine as raw pointers, so the
replica lives in HBM3E, integrates with torch's allocator, and `values` can
be viewed zero-copy as model parameters.
"""
from __future__ import annotations

from typing import Dict, List, Optional, Sequence, Union

import torch

from . import _core

CODECS = {"1bit": 0, "fp8": 1, "int4": 2}


def _percentile(xs: List[float], q: float) -> Optional[float]:
    if not xs:
        return None
    s = sorted(xs)
    i = min(int(q * len(s)), len(s) - 1)
    return s[i]


class _SharedBase:
    """Engine lifecycle + buffer management shared by SharedTensor/SharedTable."""

    def __init__(self, host: str, port: int, sizes: Sequence[int],
                 device: torch.device, codec: str = "1bit", *,
                 snapshot_join: bool = True, use_rccl: bool = True,
                 reconnect: bool = False, preserve_subtree: bool = False,
                 keepalive_s: float = 1.0,
                 bw_limit: float = 0.0, sync_interval_s: float = 0.0,
                 expected_children: int = 2,
                 provision_up: bool = True, explicit_parent: str = "",
                 listen_port: int = 0, join_timeout_s: float = 60.0,
                 rms_sample_stride: int = 1, lagged_scale: bool = False,
                 delta_dtype: torch.dtype = torch.float32,
                 use_graphs: bool = False):
        if codec not in CODECS:
            raise ValueError(f"codec must be one of {list(CODECS)}")
        self.device = torch.device(device)
        self._gpu = self.device.type == "cuda"
        cfg = _core.Config()
        cfg.host = host
        cfg.port = int(port)
        cfg.device = (self.device.index or 0) if self._gpu else -1
        cfg.codec = CODECS[codec]
        cfg.snapshot_join = snapshot_join
        cfg.use_rccl = use_rccl
        cfg.reconnect = reconnect
        if preserve_subtree and not snapshot_join:
            raise ValueError("preserve_subtree requires snapshot_join")
        cfg.preserve_subtree = preserve_subtree
        cfg.keepalive_s = keepalive_s
        cfg.bw_limit = float(bw_limit)
        cfg.min_round_interval_s = float(sync_interval_s)
        cfg.expected_children = expected_children
        cfg.sizes = [int(s) for s in sizes]
        cfg.explicit_parent = explicit_parent
        cfg.listen_port = int(listen_port)
        cfg.join_timeout_s = join_timeout_s
        cfg.rms_sample_stride = rms_sample_stride
        cfg.lagged_scale = bool(lagged_scale)
        if delta_dtype not in (torch.float32, torch.bfloat16):
            raise ValueError("delta_dtype must be float32 or bfloat16")
        if delta_dtype == torch.bfloat16 and not self._gpu:
            raise ValueError("bf16 residual deltas need a GPU engine")
        cfg.delta_bf16 = delta_dtype == torch.bfloat16
        self.delta_dtype = delta_dtype
        cfg.use_graphs = bool(use_graphs)
        self.codec = codec
        self.n = int(sum(sizes))
        self._cfg = cfg
        self._closed = False

        msg = _core.msg_bytes(cfg)
        self.values = torch.zeros(self.n, dtype=torch.float32, device=self.device)
        self._eng = _core.Engine(cfg)
        self._eng.set_values(self.values.data_ptr())

        self._link_bufs = []
        links = []
        if provision_up:
            links.append(0)
        links += [1, 2][: max(0, min(2, expected_children))]
        for li in (0, 1, 2):
            if li not in links:
                continue
            delta = torch.zeros(self.n, dtype=self.delta_dtype,
                                device=self.device)
            if self._gpu:
                # zeros (one-time cost): the scales-area padding beyond 4*T
                # bytes is never written by any kernel, and uninitialized HBM
                # must not leak onto the wire (WIRE_FORMAT.md: pad is zero)
                send_buf = torch.zeros(msg, dtype=torch.uint8, device=self.device)
                recv_buf = torch.zeros(msg, dtype=torch.uint8, device=self.device)
                send_pin = torch.empty(8 + msg, dtype=torch.uint8, pin_memory=True)
                recv_pin = torch.empty(8 + msg, dtype=torch.uint8, pin_memory=True)
                bufs = (delta, send_buf, recv_buf, send_pin, recv_pin)
                self._eng.set_link_buffers(li, delta.data_ptr(), send_buf.data_ptr(),
                                           recv_buf.data_ptr(), send_pin.data_ptr(),
                                           recv_pin.data_ptr())
            else:
                send_pin = torch.zeros(8 + msg, dtype=torch.uint8)
                recv_pin = torch.zeros(8 + msg, dtype=torch.uint8)
                bufs = (delta, send_pin, recv_pin)
                self._eng.set_link_buffers(li, delta.data_ptr(), 0, 0,
                                           send_pin.data_ptr(), recv_pin.data_ptr())
            self._link_bufs.append(bufs)

    # -- lifecycle ---------------------------------------------------------
    def _start(self):
        self._eng.start()

    @property
    def is_master(self) -> bool:
        return self._eng.is_master()

    @property
    def listen_port(self) -> int:
        return self._eng.listen_port()

    def close(self):
        if not self._closed:
            self._closed = True
            self._eng.close()

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass

    # -- shared-memory ops -------------------------------------------------
    def _stream(self) -> int:
        if self._gpu:
            return torch.cuda.current_stream(self.device).cuda_stream
        return 0

    def _add_flat(self, flat: torch.Tensor):
        if flat.dtype != torch.float32 or not flat.is_contiguous():
            raise TypeError("expected a contiguous float32 tensor")
        self._eng.add_from(flat.data_ptr(), flat.numel(), self._stream())

    def _copy_flat(self, flat: torch.Tensor):
        if flat.dtype != torch.float32 or not flat.is_contiguous():
            raise TypeError("expected a contiguous float32 tensor")
        self._eng.copy_to(flat.data_ptr(), flat.numel(), self._stream())

    def fused_sgd_step(self, momentum_buf: torch.Tensor, grad: torch.Tensor,
                       lr: float, momentum: float = 0.9):
        """m = mu*m + g; u = -lr*m; {replica, link deltas} += u — one fused
        HBM pass (HIP kernel k_fused_sgd) instead of optimizer + addFromTensor."""
        if momentum_buf.numel() != self.n or grad.numel() != self.n:
            raise ValueError("size mismatch")
        self._eng.fused_sgd(momentum_buf.data_ptr(), grad.data_ptr(),
                            float(lr), float(momentum), self._stream())

    def fused_sgd_bf16_step(self, momentum_buf: torch.Tensor,
                            grad_bf16: torch.Tensor, shadow_bf16: torch.Tensor,
                            lr: float, momentum: float = 0.9):
        """Mixed-precision fused step: bf16 grads in, fp32 master updated,
        bf16 shadow params refreshed, link deltas staged — one HBM pass
        (HIP kernel k_fused_sgd_bf16)."""
        if grad_bf16.dtype != torch.bfloat16 or shadow_bf16.dtype != torch.bfloat16:
            raise TypeError("grad/shadow must be bfloat16")
        if not (momentum_buf.numel() == grad_bf16.numel()
                == shadow_bf16.numel() == self.n):
            raise ValueError("size mismatch")
        self._eng.fused_sgd_bf16(momentum_buf.data_ptr(), grad_bf16.data_ptr(),
                                 shadow_bf16.data_ptr(), float(lr),
                                 float(momentum), self._stream())

    def fused_adamw_step(self, mom: torch.Tensor, vel: torch.Tensor,
                         grad: torch.Tensor, step: int, lr: float,
                         betas=(0.9, 0.999), eps: float = 1e-8,
                         weight_decay: float = 0.0):
        """torch.optim.AdamW-semantics update feeding the shared tensor:
        fp32 m/v, decoupled weight decay on the pre-update master weight,
        update applied to the replica AND staged into every link delta in
        one HBM pass (HIP kernel k_fused_adamw)."""
        if not (mom.numel() == vel.numel() == grad.numel() == self.n):
            raise ValueError("size mismatch")
        if grad.dtype != torch.float32:
            raise TypeError("fused_adamw_step takes fp32 grads "
                            "(use fused_adamw_bf16_step)")
        self._eng.fused_adamw(mom.data_ptr(), vel.data_ptr(), grad.data_ptr(),
                              False, 0, float(lr), float(betas[0]),
                              float(betas[1]), float(eps),
                              float(weight_decay), int(step), self._stream())

    def fused_adamw_bf16_step(self, mom: torch.Tensor, vel: torch.Tensor,
                              grad_bf16: torch.Tensor,
                              shadow_bf16: torch.Tensor, step: int, lr: float,
                              betas=(0.9, 0.999), eps: float = 1e-8,
                              weight_decay: float = 0.0):
        """Mixed-precision fused AdamW: bf16 grads in, fp32 master updated,
        bf16 shadow params refreshed (folding concurrent gossip), link
        deltas staged — one HBM pass."""
        if grad_bf16.dtype != torch.bfloat16 or shadow_bf16.dtype != torch.bfloat16:
            raise TypeError("grad/shadow must be bfloat16")
        if not (mom.numel() == vel.numel() == grad_bf16.numel()
                == shadow_bf16.numel() == self.n):
            raise ValueError("size mismatch")
        self._eng.fused_adamw(mom.data_ptr(), vel.data_ptr(),
                              grad_bf16.data_ptr(), True,
                              shadow_bf16.data_ptr(), float(lr),
                              float(betas[0]), float(betas[1]), float(eps),
                              float(weight_decay), int(step), self._stream())

    # -- observability -----------------------------------------------------
    def stats(self) -> dict:
        links = self._eng.link_stats()
        recv_scales = self._eng.recent_scales_recv()
        sent_scales = self._eng.recent_scales_sent()
        return {
            "is_master": self.is_master,
            "listen_port": self.listen_port,
            "links": links,
            "rounds_sent": sum(l["rounds_sent"] for l in links),
            "rounds_recv": sum(l["rounds_recv"] for l in links),
            "bytes_sent": sum(l["bytes_sent"] for l in links),
            "bytes_recv": sum(l["bytes_recv"] for l in links),
            # the per-round scale IS the live staleness measure: every packet
            # moves each element by exactly +-scale
            "staleness_p50": _percentile(recv_scales, 0.5),
            "staleness_p90": _percentile(recv_scales, 0.9),
            "sent_scale_p50": _percentile(sent_scales, 0.5),
            "reconnects": self._eng.reconnect_count(),
            "last_error": self._eng.last_error(),
        }

    def notify(self):
        self._eng.notify_dirty()


class SharedTensor(_SharedBase):
    """A distributed shared tensor (reference API parity).

    The first process to bind (host, port) becomes the master and seeds the
    shared state with `tensor`'s contents; later processes join the
    self-organizing binary tree and receive the state (snapshot fast-path or
    the reference's converging-delta bootstrap).
    """

    def __init__(self, host: str, port: int, tensor: torch.Tensor, **kw):
        if tensor.dtype != torch.float32:
            raise TypeError("shared tensor must be float32 (fp32 replica + "
                            "compressed wire deltas)")
        self.shape = tuple(tensor.shape)
        super().__init__(host, port, [tensor.numel()],
                         device=tensor.device, **kw)
        self._start()
        if self.is_master:
            self._add_flat(tensor.detach().contiguous().view(-1).float())

    # reference-style API -------------------------------------------------
    def copy_to_tensor(self, tensor: torch.Tensor):
        if tuple(tensor.shape) != self.shape:
            tensor = tensor.view(self.shape)
        flat = tensor.view(-1)
        if not flat.is_contiguous() or flat.dtype != torch.float32:
            raise TypeError("copy_to_tensor needs a contiguous float32 tensor")
        self._copy_flat(flat)

    def add_from_tensor(self, tensor: torch.Tensor):
        flat = tensor.detach().contiguous().view(-1)
        if flat.dtype != torch.float32:
            raise TypeError("add_from_tensor needs a float32 tensor")
        if flat.numel() != self.n:
            raise ValueError("size mismatch")
        self._add_flat(flat)

    # camelCase aliases matching the reference Lua API (example.lua:15,22)
    copyToTensor = copy_to_tensor
    addFromTensor = add_from_tensor

    def view(self) -> torch.Tensor:
        """Zero-copy view of the live replica (mutated by the gossip engine;
        reads are the async-approximate contract, README.md:20-24)."""
        return self.values.view(self.shape)

    # checkpointing: the reference's only "restore" is joining (the
    # accumulated-delta bootstrap); we add explicit save/restore on top
    def save(self, path: str):
        torch.save({"values": self.values.detach().cpu(),
                    "shape": self.shape, "codec": self.codec}, path)

    @classmethod
    def restore(cls, host: str, port: int, path: str,
                device="cpu", **kw) -> "SharedTensor":
        """Create/join a shared tensor seeded from a checkpoint.  If this
        process becomes master, the checkpointed values seed the tree;
        otherwise the live tree state wins (by reference join semantics)."""
        ckpt = torch.load(path, map_location="cpu", weights_only=True)
        seed = ckpt["values"].view(ckpt["shape"]).to(device)
        return cls(host, port, seed, **kw)


class SharedTable(_SharedBase):
    """Table-of-tensors sync with per-tensor scales (reference README.md:41).

    Shares a dict/list of named tensors as one engine instance; each packet
    carries one scale per member tensor, so small and large tensors converge
    at their own magnitudes.
    """

    def __init__(self, host: str, port: int,
                 tensors: Union[Dict[str, torch.Tensor], Sequence[torch.Tensor]],
                 **kw):
        if isinstance(tensors, dict):
            items = list(tensors.items())
        else:
            items = [(str(i), t) for i, t in enumerate(tensors)]
        if not items:
            raise ValueError("empty table")
        dev = items[0][1].device
        for k, t in items:
            if t.device != dev:
                raise ValueError("all table tensors must share a device")
            if t.dtype != torch.float32:
                raise TypeError(f"table tensor {k} must be float32")
        self.names = [k for k, _ in items]
        self.shapes = {k: tuple(t.shape) for k, t in items}
        sizes = [t.numel() for _, t in items]
        super().__init__(host, port, sizes, device=dev, **kw)
        self._offsets = {}
        off = 0
        for (k, t), s in zip(items, sizes):
            self._offsets[k] = (off, off + s)
            off += s
        self._start()
        if self.is_master:
            flat = torch.cat([t.detach().contiguous().view(-1) for _, t in items])
            self._add_flat(flat)

    def tensor_view(self, name: str) -> torch.Tensor:
        a, b = self._offsets[name]
        return self.values[a:b].view(self.shapes[name])

    def views(self) -> Dict[str, torch.Tensor]:
        return {k: self.tensor_view(k) for k in self.names}

    def add_from_tensors(self, tensors: Union[Dict[str, torch.Tensor], Sequence[torch.Tensor], torch.Tensor]):
        if isinstance(tensors, torch.Tensor):
            flat = tensors.detach().contiguous().view(-1)
        elif isinstance(tensors, dict):
            flat = torch.cat([tensors[k].detach().contiguous().view(-1) for k in self.names])
        else:
            flat = torch.cat([t.detach().contiguous().view(-1) for t in tensors])
        if flat.numel() != self.n:
            raise ValueError("size mismatch")
        self._add_flat(flat.float())

    def copy_to_tensors(self, out: Optional[Dict[str, torch.Tensor]] = None) -> Dict[str, torch.Tensor]:
        if out is None:
            out = {k: torch.empty(self.shapes[k], dtype=torch.float32,
                                  device=self.device) for k in self.names}
        for k in self.names:
            a, b = self._offsets[k]
            out[k].view(-1).copy_(self.values[a:b])
        return out

    @classmethod
    def from_module(cls, host: str, port: int, module: torch.nn.Module, **kw):
        params = {n: p.data for n, p in module.named_parameters()}
        return cls(host, port, params, **kw)


# public name for the flat multi-tensor-capable base (bench/paramsync use it
# directly when no shape semantics are needed)
SharedFlat = _SharedBase


def create_or_fetch(host: str, port: int, tensor: torch.Tensor, **kw) -> SharedTensor:
    """Reference entrypoint parity (sharedtensor.createOrFetch,
    sharedtensor.c:347-391)."""
    return SharedTensor(host, port, tensor, **kw)


createOrFetch = create_or_fetch
