"""Op graph DSL (parity: python/scannerpy/op.py).

`sc.ops.<Name>(input_col=..., arg=..., device=...)` materializes an Op node;
`@scanner_amd.register_python_op()` registers a Python function/class as an
op, deriving input/output columns from type annotations (FrameType ->
frame column, bytes -> blob column, Sequence[...] -> batched,
nested Sequence for stencil)."""
import inspect
import typing

import msgpack

from .common import ColumnType, DeviceType, FrameType, ScannerException


class OpColumn:
    def __init__(self, op, name, typ):
        self.op = op
        self.name = name
        self.type = typ

    def compress(self, **kw):
        """Mark a frame column for codec compression at the sink (parity:
        OpColumn.compress_video op.py:47-106)."""
        self.op._compress[self.name] = dict(kw)
        return self

    def compress_video(self, **kw):
        return self.compress(codec="svc", **kw)

    def lossless(self):
        return self.compress(codec="svc")


class Op:
    def __init__(self, client, name, inputs, args=None, device=DeviceType.CPU,
                 batch=0, stencil=None, warmup=-1, stream_args=None,
                 output_columns=None):
        self._client = client
        self._name = name
        self._inputs = inputs              # list of OpColumn
        self._args = args or {}
        self._device = device
        self._batch = batch
        self._stencil = stencil or []
        self._warmup = warmup
        self._stream_args = stream_args    # per-stream args (list) or None
        self._compress = {}
        if output_columns is None:
            info = client._op_info(name)
            output_columns = [(n, ColumnType(t))
                              for n, t in info["output_columns"]]
        self._outputs = [OpColumn(self, n, t) for n, t in output_columns]

    def outputs(self):
        return list(self._outputs)

    # Single-output convenience: compression annotations apply to the sole
    # output column (reference ops hand back OpColumns directly).
    def compress(self, **kw):
        return self._single().compress(**kw)

    def compress_video(self, **kw):
        return self._single().compress_video(**kw)

    def lossless(self):
        return self._single().lossless()

    def __getitem__(self, name):
        for c in self._outputs:
            if c.name == name:
                return c
        raise ScannerException(f"op {self._name} has no output column {name}")

    def _single(self):
        if len(self._outputs) != 1:
            raise ScannerException(
                f"op {self._name} has {len(self._outputs)} outputs; "
                "select one explicitly")
        return self._outputs[0]


def _as_column(x):
    if isinstance(x, OpColumn):
        return x
    if isinstance(x, Op):
        return x._single()
    raise ScannerException(f"expected an op output column, got {type(x)}")


class OpGenerator:
    """`sc.ops.<OpName>(...)` (parity: OpGenerator op.py:121)."""

    def __init__(self, client):
        self._client = client

    def __getattr__(self, name):
        if name.startswith("_"):
            raise AttributeError(name)
        client = self._client

        def make(*, device=DeviceType.CPU, batch=0, stencil=None,
                 warmup=-1, args=None, stream_args=None, **kwargs):
            info = client._op_info(name)
            in_cols = [n for n, _ in info["input_columns"]]
            inputs = []
            op_args = dict(args or {})
            if info["variadic"]:
                cols = kwargs.pop("inputs")
                inputs = [_as_column(c) for c in cols]
                op_args.update(kwargs)
            else:
                for cn in in_cols:
                    if cn not in kwargs:
                        raise ScannerException(
                            f"op {name} missing input column '{cn}'")
                    inputs.append(_as_column(kwargs.pop(cn)))
                # remaining kwargs are op args
                op_args.update(kwargs)
            return Op(client, name, inputs, args=op_args, device=device,
                      batch=batch, stencil=stencil, warmup=warmup,
                      stream_args=stream_args)

        return make


# ---------------- @register_python_op ----------------

_FRAME_ANN = ("FrameType", FrameType)


def _ann_to_column(ann):
    """Returns (is_frame, seq_depth) for an annotation. Depth 1 means a
    batch (or, when the op declares stencil=, the stencil window); depth 2
    means stencil-within-batch (parity: reference op.py:389-535)."""
    import collections.abc
    origin = typing.get_origin(ann)
    if origin in (list, collections.abc.Sequence):
        inner = typing.get_args(ann)[0]
        f, d = _ann_to_column(inner)
        return f, d + 1
    if ann is FrameType or ann == "FrameType":
        return True, 0
    return False, 0


# name -> cloudpickled re-registration thunk, for shipping python ops to
# worker processes (parity: RegisterOp/RegisterPythonKernel broadcast,
# master.cpp:751-813)
_PY_OP_PICKLES = {}


def register_python_op(name=None, device_type=DeviceType.CPU, batch=0,
                       stencil=None, bounded_state=False, warmup=0,
                       unbounded_state=False, isolation="thread"):
    """Decorator registering a Python function or Kernel class as an op
    (parity: @scannerpy.register_python_op op.py:317). Input/output columns
    derive from the annotations of `execute` (class) or the function.

    isolation="process" runs each kernel instance in its own child Python
    process connected by pipes (parity: the reference's PythonKernel
    subprocess model, python_kernel.cpp:30-103): N pipeline instances of a
    CPU-heavy Python op then compute on N cores instead of serializing on
    this process's GIL. The parent blocks in pipe reads (GIL released)
    while children work; inputs/outputs cross the pipe pickled. Default
    "thread" runs in-process (no marshaling cost — right for cheap ops)."""

    def deco(fn_or_cls):
        from . import _core
        opname = name or fn_or_cls.__name__
        # Refuse to shadow a first-party C++ op (a python op silently
        # replacing Histogram etc. breaks every pipeline using it);
        # RE-registering a python op of the same name stays allowed —
        # notebook-style iteration depends on it.
        if (opname in _core.registered_ops()
                and opname not in _PY_OP_PICKLES):
            raise ScannerException(
                f"op '{opname}' is already a registered C++ op; "
                "pick a different name")
        is_cls = inspect.isclass(fn_or_cls)
        target = fn_or_cls.execute if is_cls else fn_or_cls
        sig = inspect.signature(target)
        params = [p for p in sig.parameters.values()
                  if p.name not in ("self", "config")]
        in_cols = []
        stenciled = stencil is not None
        max_depth = 0
        for p in params:
            if p.annotation is inspect.Parameter.empty:
                raise ScannerException(
                    f"python op {opname}: parameter {p.name} needs a type "
                    "annotation")
            f, d = _ann_to_column(p.annotation)
            max_depth = max(max_depth, d)
            in_cols.append((p.name, 1 if f else 0))
        # depth 1 = batch, unless a stencil is declared (then it is the
        # stencil window); depth 2 = stencil within batch
        batched = max_depth >= 2 or (max_depth == 1 and not stenciled)
        ret = sig.return_annotation
        out_cols = []
        if typing.get_origin(ret) is tuple:
            for i, r in enumerate(typing.get_args(ret)):
                f, _ = _ann_to_column(r)
                out_cols.append((f"out{i}", 1 if f else 0))
        else:
            f, _ = _ann_to_column(ret)
            out_cols.append(("out", 1 if f else 0))

        the_stencil = list(stencil) if stencil else [0]
        eff_batch = batch if batch > 0 else (1024 if batched else 1)

        def factory(args_bytes):
            if isolation == "process":
                import cloudpickle
                payload = cloudpickle.dumps(
                    (fn_or_cls, is_cls, args_bytes, batched, stenciled,
                     len(in_cols), out_cols))
                return _SubprocessKernel(payload)
            return _PyKernelAdapter(fn_or_cls, is_cls, args_bytes, batched,
                                    stenciled, len(in_cols), out_cols)

        _core.register_python_op(
            opname, factory, in_cols, out_cols, int(device_type),
            eff_batch, the_stencil, bounded_state, warmup, unbounded_state)
        fn_or_cls._scanner_op_name = opname
        try:
            import cloudpickle

            def _remote_register(fn_or_cls=fn_or_cls, kw=dict(
                    name=opname, device_type=device_type, batch=batch,
                    stencil=stencil, bounded_state=bounded_state,
                    warmup=warmup, unbounded_state=unbounded_state)):
                from scanner_amd.op import register_python_op as rpo
                rpo(**kw)(fn_or_cls)

            _PY_OP_PICKLES[opname] = cloudpickle.dumps(_remote_register)
        except Exception:
            pass  # op still usable locally
        return fn_or_cls

    return deco


class Kernel:
    """Base class for class-style Python ops (parity: scannerpy.Kernel)."""

    def __init__(self, config):
        self.config = config

    def new_stream(self, args):
        pass

    def reset(self):
        pass

    def fetch_resources(self, args):
        pass

    def setup_with_resources(self, args):
        pass

    def execute(self, *cols):
        raise NotImplementedError


class KernelConfigPy:
    def __init__(self, args):
        self.args = args


class _PyKernelAdapter:
    """Adapts user fn/class to the raw C++ interface
    (cols[in][row][stencil] -> [out][row])."""

    def __init__(self, fn_or_cls, is_cls, args_bytes, batched, stenciled,
                 n_in, out_cols):
        args = msgpack.unpackb(args_bytes) if args_bytes else {}
        self._batched = batched
        self._stenciled = stenciled
        self._n_out = len(out_cols)
        if is_cls:
            self._obj = fn_or_cls(KernelConfigPy(args), **args)
            self._fn = self._obj.execute
        else:
            self._obj = None
            kw = args

            def call(*cols):
                return fn_or_cls(*cols, **kw)

            self._fn = call

    def new_stream(self, args_bytes):
        if self._obj is not None and args_bytes:
            args = msgpack.unpackb(args_bytes)
            self._obj.new_stream(**args) if isinstance(args, dict) \
                else self._obj.new_stream(args)

    def reset(self):
        if self._obj is not None:
            self._obj.reset()

    def fetch_resources(self, args_bytes):
        if self._obj is not None:
            args = msgpack.unpackb(args_bytes) if args_bytes else {}
            self._obj.fetch_resources(args)

    def setup_with_resources(self, args_bytes):
        if self._obj is not None and hasattr(self._obj,
                                             "setup_with_resources"):
            args = msgpack.unpackb(args_bytes) if args_bytes else {}
            self._obj.setup_with_resources(args)

    def execute(self, cols):
        n_rows = len(cols[0]) if cols else 0
        outs = [[] for _ in range(self._n_out)]
        if self._batched:
            ins = []
            for col in cols:
                if self._stenciled:
                    ins.append([row for row in col])
                else:
                    ins.append([row[0] for row in col])
            result = self._fn(*ins)
            if self._n_out == 1:
                result = (result,)
            for c in range(self._n_out):
                outs[c] = list(result[c])
        else:
            for r in range(n_rows):
                ins = []
                for col in cols:
                    win = col[r]
                    ins.append(win if self._stenciled else win[0])
                result = self._fn(*ins)
                if self._n_out == 1:
                    result = (result,)
                for c in range(self._n_out):
                    outs[c].append(result[c])
        return outs


# ---- subprocess kernel isolation (parity: PythonKernel child process,
# reference python_kernel.cpp:30-103 + kernel.py python_kernel_fn) ----

def _pipe_send(f, obj):
    import pickle
    import struct
    data = pickle.dumps(obj, protocol=pickle.HIGHEST_PROTOCOL)
    f.write(struct.pack("<Q", len(data)))
    f.write(data)
    f.flush()


def _pipe_recv(f):
    import pickle
    import struct
    hdr = f.read(8)
    if len(hdr) < 8:
        raise EOFError("kernel subprocess pipe closed")
    (n,) = struct.unpack("<Q", hdr)
    return pickle.loads(f.read(n))


def _child_main():
    """Entry point of a kernel child process: builds the adapter from the
    cloudpickled payload, then serves method calls over stdin/stdout."""
    import sys

    import cloudpickle
    inp = sys.stdin.buffer
    outp = sys.stdout.buffer
    # anything the user op prints must not corrupt the pipe
    sys.stdout = sys.stderr
    kind, payload = _pipe_recv(inp)
    assert kind == "init"
    adapter = _PyKernelAdapter(*cloudpickle.loads(payload))
    _pipe_send(outp, ("ok", None))
    while True:
        try:
            msg = _pipe_recv(inp)
        except EOFError:
            return
        kind = msg[0]
        if kind == "exit":
            return
        try:
            result = getattr(adapter, kind)(*msg[1:])
            _pipe_send(outp, ("ok", result))
        except Exception as e:
            import traceback
            _pipe_send(outp, ("err", f"{type(e).__name__}: {e}\n"
                              + traceback.format_exc()))


class _SubprocessKernel:
    """Runs a _PyKernelAdapter in a child process; this proxy mirrors its
    interface for the C++ bridge. The parent blocks in pipe reads — the
    GIL is released during the read syscall — so N pipeline instances of a
    CPU-heavy kernel genuinely use N cores."""

    def __init__(self, payload):
        import os
        import subprocess
        import sys
        env = dict(os.environ)
        repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
        env["PYTHONPATH"] = repo + os.pathsep + env.get("PYTHONPATH", "")
        self._p = subprocess.Popen(
            [sys.executable, "-c",
             "from scanner_amd.op import _child_main; _child_main()"],
            stdin=subprocess.PIPE, stdout=subprocess.PIPE, env=env)
        self._call_raw("init", payload)

    def _call_raw(self, kind, *args):
        if self._p.poll() is not None:
            raise ScannerException("kernel subprocess died "
                                   f"(rc={self._p.returncode})")
        _pipe_send(self._p.stdin, (kind,) + args)
        status, value = _pipe_recv(self._p.stdout)
        if status == "err":
            raise ScannerException(f"python kernel (subprocess): {value}")
        return value

    def new_stream(self, args_bytes):
        return self._call_raw("new_stream", args_bytes)

    def reset(self):
        return self._call_raw("reset")

    def fetch_resources(self, args_bytes):
        return self._call_raw("fetch_resources", args_bytes)

    def setup_with_resources(self, args_bytes):
        return self._call_raw("setup_with_resources", args_bytes)

    def execute(self, cols):
        # numpy views from C++ pickle as copies — exactly the marshaling
        # the reference pays on its pipes too
        return self._call_raw("execute", cols)

    def close(self):
        p = getattr(self, "_p", None)
        if p is None or p.poll() is not None:
            return
        try:
            _pipe_send(p.stdin, ("exit",))
            p.stdin.close()
            p.wait(timeout=5)
        except Exception:
            p.kill()

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass
