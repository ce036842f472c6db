"""Control-plane RPC: grpcio with msgpack-serialized messages.

Parity role: the reference's gRPC control plane (rpc.proto). This image has
no protoc, so instead of generated stubs we use grpc generic handlers with
msgpack payloads — same wire properties (HTTP/2, deadlines, errors), no
codegen."""
import threading
from concurrent import futures

import grpc
import msgpack


def _ser(obj):
    return msgpack.packb(obj, use_bin_type=True)


def _des(b):
    return msgpack.unpackb(b, raw=False)


class RpcError(Exception):
    pass


class RpcServer:
    """handlers: {method_name: fn(request_dict) -> response_dict}"""

    def __init__(self, addr, handlers, max_workers=16):
        self._server = grpc.server(
            futures.ThreadPoolExecutor(max_workers=max_workers),
            options=[("grpc.max_receive_message_length", 1 << 30),
                     ("grpc.max_send_message_length", 1 << 30)])
        method_handlers = {}
        for name, fn in handlers.items():
            method_handlers[name] = grpc.unary_unary_rpc_method_handler(
                self._wrap(fn), request_deserializer=_des,
                response_serializer=_ser)
        self._server.add_generic_rpc_handlers(
            (grpc.method_handlers_generic_handler("scanner",
                                                  method_handlers),))
        self.port = self._server.add_insecure_port(addr)
        if self.port == 0:
            raise RpcError(f"could not bind {addr}")
        self._server.start()

    @staticmethod
    def _wrap(fn):
        def h(req, ctx):
            try:
                return fn(req or {})
            except Exception as e:  # surfaced as grpc error details
                ctx.abort(grpc.StatusCode.INTERNAL,
                          f"{type(e).__name__}: {e}")
        return h

    def stop(self, grace=0.5):
        self._server.stop(grace)

    def wait(self):
        self._server.wait_for_termination()


class RpcClient:
    def __init__(self, addr):
        self.addr = addr
        self._ch = grpc.insecure_channel(
            addr, options=[("grpc.max_receive_message_length", 1 << 30),
                           ("grpc.max_send_message_length", 1 << 30)])
        self._lock = threading.Lock()
        self._methods = {}

    def call(self, method, payload=None, timeout=60):
        with self._lock:
            if method not in self._methods:
                self._methods[method] = self._ch.unary_unary(
                    f"/scanner/{method}", request_serializer=_ser,
                    response_deserializer=_des)
            fn = self._methods[method]
        try:
            return fn(payload or {}, timeout=timeout)
        except grpc.RpcError as e:
            raise RpcError(f"{method} -> {e.code().name}: {e.details()}")

    def try_call(self, method, payload=None, timeout=10):
        try:
            return self.call(method, payload, timeout)
        except RpcError:
            return None

    def close(self):
        self._ch.close()
