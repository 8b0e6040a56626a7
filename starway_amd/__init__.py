"""starway_amd — MI355X-native async zero-copy tagged messaging for Python.

A brand-new implementation with the capability surface of Clouder0/starway
(reference: /root/reference/src/starway/__init__.py — API contract only).
Where the reference wrapped OpenUCX, this library ships its own transport:

* TCP control + CPU data plane with UCX-style eager semantics
* GPU data plane: tagged send/recv of HIP device tensors, zero-copy over
  xGMI via hipIpc handle exchange and hand-written gfx950 copy kernels
* completion loop: per-object C++ progress thread polling hipEvents, with
  asyncio bridging via ``loop.call_soon_threadsafe``

Public surface (parity with the reference): ``Server``, ``Client``,
``ServerEndpoint``, ``check_sys_libs``, ``list_benchmark_scenarios``.
"""
from __future__ import annotations

import asyncio
import ctypes
import os
import weakref
from collections import deque
from collections.abc import Callable
from importlib.util import find_spec
from pathlib import Path
from typing import Any, Literal

# ---------------------------------------------------------------------------
# HIP runtime loader policy (the analog of the reference's UCX loader,
# reference src/starway/__init__.py:14-51): our native core NEEDs
# libamdhip64.so.7. PyTorch-ROCm bundles its own copy of the runtime with
# the same soname; if _core loads the system ROCm runtime first, a later
# torch import binds to it too and mixed-version HSA state breaks device
# discovery ("No HIP GPUs are available"). Default policy: when torch is
# installed, preload torch's bundled runtime (RTLD_GLOBAL) so everyone
# shares one copy. STARWAY_USE_SYSTEM_HIP=true opts into the system ROCm
# runtime instead (only safe in torch-free processes).
# ---------------------------------------------------------------------------
# The engine uses several concurrent streams per device (pull lanes +
# small-message push/unpack/doorbell lanes). ROCm multiplexes streams onto
# GPU_MAX_HW_QUEUES hardware queues (default 4); once oversubscribed,
# co-mapped streams are time-sliced at ~ms granularity and a resident
# doorbell kernel can starve a push stream (measured: bistable 3.3 ms
# pingpong RTT). Raise the default before the HIP runtime initializes;
# a user-set value is respected.
os.environ.setdefault("GPU_MAX_HW_QUEUES", "16")

_used_hip = "system"
if os.environ.get("STARWAY_USE_SYSTEM_HIP", "false") != "true":
    _torch_spec = find_spec("torch")
    if _torch_spec and _torch_spec.origin:
        _libdir = Path(_torch_spec.origin).parent / "lib"
        for _name in ("libhsa-runtime64.so", "libamdhip64.so"):
            _p = _libdir / _name
            if _p.exists():
                try:
                    ctypes.CDLL(str(_p), mode=ctypes.RTLD_GLOBAL)
                    _used_hip = "wheel"
                except OSError:
                    pass

try:
    from ._core import Client as _Client
    from ._core import Context, ServerEndpoint
    from ._core import Server as _Server
    from ._core import gpu_available, gpu_device_count, ipc_invalidate
except ImportError as exc:  # pragma: no cover - build guidance
    raise ImportError(
        "starway_amd._core is not built. Run `python build_ext.py` (or "
        "`python -c 'import __graft_entry__ as g; g.build()'`) from the repo "
        "root first."
    ) from exc

from .benchmarks import list_scenarios as list_benchmark_scenarios  # noqa: E402


def check_sys_libs() -> Literal["system"] | Literal["wheel"]:
    """Which HIP runtime the native core is bound to: "wheel" = the copy
    bundled inside the installed torch package (default when torch is
    present), "system" = /opt/rocm. The reference's analog reported which
    UCX .so was loaded (reference src/starway/__init__.py:63-65)."""
    return _used_hip  # type: ignore[return-value]


_context = Context()

# IPC hygiene: imported hipIpc mappings are cached for the process lifetime
# (registration-cache pattern, safe under torch's caching allocator). Close
# them at interpreter exit; call ipc_invalidate() yourself after returning
# GPU memory to the driver mid-run (e.g. torch.cuda.empty_cache()).
if gpu_available():
    import atexit

    atexit.register(ipc_invalidate)


class _CudaShim:
    """Zero-copy view of a torch HIP tensor for the native core.

    Exposes __cuda_array_interface__ built from data_ptr()/nbytes, pinning
    the tensor via a strong reference (the core keeps this shim alive as the
    op's keepalive). Used instead of the tensor's own CAI so behavior is
    identical across torch builds.
    """

    __slots__ = ("_t", "__cuda_array_interface__")

    def __init__(self, t) -> None:
        es = t.element_size()
        if t.is_contiguous():
            nbytes = t.numel() * es
            self.__cuda_array_interface__ = {
                "data": (t.data_ptr(), False),
                "shape": (nbytes,),
                "typestr": "|u1",
                "strides": None,
                "version": 2,
            }
        elif t.dim() == 2 and t.stride(1) == 1:
            # 2D row-strided slice: moved natively by the strided
            # pack/unpack kernels — no .contiguous() staging copy.
            r, c = t.shape
            self.__cuda_array_interface__ = {
                "data": (t.data_ptr(), False),
                "shape": (r, c * es),
                "typestr": "|u1",
                "strides": (t.stride(0) * es, 1),
                "version": 2,
            }
        else:
            raise ValueError(
                "device message buffers must be contiguous or 2D row-strided "
                "(call .contiguous() first)"
            )
        self._t = t
        try:
            t._sw_cai = self.__cuda_array_interface__
        except Exception:
            pass  # tensor subclass without attribute support


def _norm_buffer(buf: Any) -> Any:
    """Normalize a message buffer for the native core.

    Accepts numpy arrays / anything with the buffer protocol (host path) and
    torch tensors: HIP tensors go zero-copy via a __cuda_array_interface__
    shim; CPU torch tensors are viewed as numpy (zero-copy).
    """
    mod = type(buf).__module__
    if mod.startswith("torch"):
        if buf.is_cuda:
            # The interface dict is cached on the tensor (hot benches reuse
            # message buffers); the shim itself is rebuilt per op because
            # it doubles as the op's keepalive (a cached shim would form a
            # tensor<->shim cycle and pin GPU memory on the GC).
            cai = getattr(buf, "_sw_cai", None)
            if cai is None or cai["data"][0] != buf.data_ptr():
                return _CudaShim(buf)
            shim = _CudaShim.__new__(_CudaShim)
            shim.__cuda_array_interface__ = cai
            shim._t = buf
            return shim
        return buf.numpy()
    return buf


class Server:
    def __init__(self) -> None:
        self._server = _Server(_context)

    def listen(self, addr: str, port: int) -> None:
        self._server.listen(addr, port)

    def listen_address(self) -> bytes:
        self._server.listen_address()
        return self.get_worker_address()

    def set_accept_cb(self, on_accept: Callable[[ServerEndpoint], None]) -> None:
        self._server.set_accept_callback(on_accept)

    def get_worker_address(self) -> bytes:
        return self._server.get_worker_address()

    def list_clients(self):
        return self._server.list_clients()

    def aclose(self, loop: asyncio.AbstractEventLoop | None = None):
        if loop is None:
            loop = asyncio.get_running_loop()
        ret: asyncio.Future[None] = asyncio.Future(loop=loop)

        def close_cb() -> None:
            loop.call_soon_threadsafe(ret.set_result, None)

        self._server.close(close_cb)
        return ret

    # -- raw callback API ---------------------------------------------------

    def send(self, client_ep, buffer, tag, done_callback, fail_callback):
        return self._server.send(
            client_ep, _norm_buffer(buffer), tag, done_callback, fail_callback
        )

    def recv(self, buffer, tag, tag_mask, done_callback, fail_callback):
        return self._server.recv(
            _norm_buffer(buffer), tag, tag_mask, done_callback, fail_callback
        )

    def flush(self, done_callback, fail_callback):
        return self._server.flush(done_callback, fail_callback)

    def flush_ep(self, client_ep, done_callback, fail_callback):
        return self._server.flush_ep(client_ep, done_callback, fail_callback)

    # -- asyncio API --------------------------------------------------------

    def asend(self, client_ep, buffer, tag: int,
              loop: asyncio.AbstractEventLoop | None = None):
        if loop is None:
            loop = asyncio.get_running_loop()
        ret: asyncio.Future[None] = asyncio.Future(loop=loop)

        def ok() -> None:
            _disp(ret.get_loop()).post(_set_result, ret, None)

        def bad(reason: str) -> None:
            _disp(ret.get_loop()).post(_set_exception, ret, reason)

        self._server.send(client_ep, _norm_buffer(buffer), tag, ok, bad)
        return ret

    def arecv(self, buffer, tag: int, tag_mask: int,
              loop: asyncio.AbstractEventLoop | None = None):
        if loop is None:
            loop = asyncio.get_running_loop()
        ret: asyncio.Future[tuple[int, int]] = asyncio.Future(loop=loop)

        def ok(sender_tag: int, length: int) -> None:
            _disp(ret.get_loop()).post(_set_result, ret, (sender_tag, length))

        def bad(reason: str) -> None:
            _disp(ret.get_loop()).post(_set_exception, ret, reason)

        self._server.recv(_norm_buffer(buffer), tag, tag_mask, ok, bad)
        return ret

    def aflush(self, loop: asyncio.AbstractEventLoop | None = None):
        if loop is None:
            loop = asyncio.get_running_loop()
        ret: asyncio.Future[None] = asyncio.Future(loop=loop)

        def ok() -> None:
            _disp(ret.get_loop()).post(_set_result, ret, None)

        def bad(reason: str) -> None:
            _disp(ret.get_loop()).post(_set_exception, ret, reason)

        self._server.flush(ok, bad)
        return ret

    def aflush_ep(self, client_ep,
                  loop: asyncio.AbstractEventLoop | None = None):
        if loop is None:
            loop = asyncio.get_running_loop()
        ret: asyncio.Future[None] = asyncio.Future(loop=loop)

        def ok() -> None:
            _disp(ret.get_loop()).post(_set_result, ret, None)

        def bad(reason: str) -> None:
            _disp(ret.get_loop()).post(_set_exception, ret, reason)

        self._server.flush_ep(client_ep, ok, bad)
        return ret

    def evaluate_perf(self, client_ep, msg_size: int) -> float:
        return self._server.evaluate_perf(client_ep, msg_size)


class Client:
    def __init__(self) -> None:
        self._client = _Client(_context)

    def aconnect(self, addr: str, port: int,
                 loop: asyncio.AbstractEventLoop | None = None):
        if loop is None:
            loop = asyncio.get_running_loop()
        ret: asyncio.Future[None] = asyncio.Future(loop=loop)

        def connection_cb(status: str) -> None:
            if status == "":
                loop.call_soon_threadsafe(_set_result, ret, None)
            else:
                loop.call_soon_threadsafe(_set_exception, ret, status)

        self._client.connect(addr, port, connection_cb)
        return ret

    def aconnect_address(self, remote_address: bytes,
                         loop: asyncio.AbstractEventLoop | None = None):
        if loop is None:
            loop = asyncio.get_running_loop()
        ret: asyncio.Future[None] = asyncio.Future(loop=loop)

        def connection_cb(status: str) -> None:
            if status == "":
                loop.call_soon_threadsafe(_set_result, ret, None)
            else:
                loop.call_soon_threadsafe(_set_exception, ret, status)

        self._client.connect_address(remote_address, connection_cb)
        return ret

    def get_worker_address(self) -> bytes:
        return self._client.get_worker_address()

    def aclose(self, loop: asyncio.AbstractEventLoop | None = None):
        if loop is None:
            loop = asyncio.get_running_loop()
        ret: asyncio.Future[None] = asyncio.Future(loop=loop)

        def close_cb() -> None:
            loop.call_soon_threadsafe(_set_result, ret, None)

        self._client.close(close_cb)
        return ret

    # -- raw callback API ---------------------------------------------------

    def send(self, buffer, tag, done_callback, fail_callback):
        return self._client.send(
            _norm_buffer(buffer), tag, done_callback, fail_callback
        )

    def recv(self, buffer, tag, tag_mask, done_callback, fail_callback):
        return self._client.recv(
            _norm_buffer(buffer), tag, tag_mask, done_callback, fail_callback
        )

    def flush(self, done_callback, fail_callback):
        return self._client.flush(done_callback, fail_callback)

    # -- asyncio API --------------------------------------------------------

    def asend(self, buffer, tag: int,
              loop: asyncio.AbstractEventLoop | None = None):
        if loop is None:
            loop = asyncio.get_running_loop()
        ret: asyncio.Future[None] = asyncio.Future(loop=loop)

        def ok() -> None:
            _disp(ret.get_loop()).post(_set_result, ret, None)

        def bad(reason: str) -> None:
            _disp(ret.get_loop()).post(_set_exception, ret, reason)

        self._client.send(_norm_buffer(buffer), tag, ok, bad)
        return ret

    def arecv(self, buffer, tag: int, tag_mask: int,
              loop: asyncio.AbstractEventLoop | None = None):
        if loop is None:
            loop = asyncio.get_running_loop()
        ret: asyncio.Future[tuple[int, int]] = asyncio.Future(loop=loop)

        def ok(sender_tag: int, length: int) -> None:
            _disp(ret.get_loop()).post(_set_result, ret, (sender_tag, length))

        def bad(reason: str) -> None:
            _disp(ret.get_loop()).post(_set_exception, ret, reason)

        self._client.recv(_norm_buffer(buffer), tag, tag_mask, ok, bad)
        return ret

    def aflush(self, loop: asyncio.AbstractEventLoop | None = None):
        if loop is None:
            loop = asyncio.get_running_loop()
        ret: asyncio.Future[None] = asyncio.Future(loop=loop)

        def ok() -> None:
            _disp(ret.get_loop()).post(_set_result, ret, None)

        def bad(reason: str) -> None:
            _disp(ret.get_loop()).post(_set_exception, ret, reason)

        self._client.flush(ok, bad)
        return ret

    def evaluate_perf(self, msg_size: int) -> float:
        return self._client.evaluate_perf(msg_size)



class _LoopDispatcher:
    """Coalesced cross-thread completion delivery.

    The engine thread fires op callbacks under the GIL; scheduling each one
    with loop.call_soon_threadsafe costs a self-pipe write + one loop
    callback PER OP. Under a completion burst (64 concurrent small
    messages) this dispatcher queues the completions and schedules ONE
    wakeup for the whole batch. All state is GIL-serialized (both the
    engine thread and the loop thread hold the GIL when touching it).
    """

    __slots__ = ("_loop", "_q", "_armed")

    def __init__(self, loop: asyncio.AbstractEventLoop) -> None:
        self._loop = loop
        self._q: deque = deque()
        self._armed = False

    def post(self, fn, *args) -> None:
        self._q.append((fn, args))
        if not self._armed:
            self._armed = True
            self._loop.call_soon_threadsafe(self._drain)

    def _drain(self) -> None:
        self._armed = False
        q = self._q
        while q:
            fn, args = q.popleft()
            fn(*args)


_dispatchers: "weakref.WeakKeyDictionary" = weakref.WeakKeyDictionary()


def _disp(loop: asyncio.AbstractEventLoop) -> _LoopDispatcher:
    d = _dispatchers.get(loop)
    if d is None:
        d = _dispatchers[loop] = _LoopDispatcher(loop)
    return d


def _set_result(fut: asyncio.Future, value) -> None:
    if not fut.done():
        fut.set_result(value)


def _set_exception(fut: asyncio.Future, reason: str) -> None:
    if not fut.done():
        fut.set_exception(Exception(reason))


__all__ = [
    "Server",
    "Client",
    "ServerEndpoint",
    "Context",
    "check_sys_libs",
    "list_benchmark_scenarios",
    "gpu_available",
    "gpu_device_count",
    "ipc_invalidate",
]
