"""gRPC server hosting the TepdistService.

The reference's grpc_service_gpu main (rpc/grpc_service_gpu.cc:16-81:
--ip --port --task_index, max message size raised, blocking wait) — here a
generic-handler gRPC server (method registry, msgpack payloads, message
size INT_MAX like the reference's SetMaxReceiveMessageSize)."""

from __future__ import annotations

import argparse
from concurrent import futures

import grpc

from tepdist_amd.rpc.service import TepdistService
from tepdist_amd.rpc.wire import pack, unpack

SERVICE_NAME = "tepdist.XlaService"

METHODS = {
    "BuildExecutionPlan": "build_execution_plan",
    "ExecutePlan": "execute_plan",
    "TransferToServerHost": "transfer_to_server_host",
    "TransferHostRawData": "transfer_host_raw_data",
    "TransferVarArgMap": "transfer_var_arg_map",
    "FetchResourceVars": "fetch_resource_vars",
    "TransferModuleAndDefCtx": "transfer_module_and_defctx",
    "DispatchPlan": "dispatch_plan",
    "InitRemoteComm": "init_remote_comm",
    "ExecuteRemotePlan": "execute_remote_plan",
    "DoRemoteSave": "do_remote_save",
    "DoRemoteRestore": "do_remote_restore",
}


class _Handler(grpc.GenericRpcHandler):
    def __init__(self, service: TepdistService):
        self.svc = service

    def service_name(self):
        return SERVICE_NAME

    def service(self, handler_call_details):
        method = handler_call_details.method.rsplit("/", 1)[-1]
        attr = METHODS.get(method)
        if attr is None:
            return None
        fn = getattr(self.svc, attr)

        def unary(request: bytes, context):
            try:
                return pack(fn(unpack(request)))
            except Exception as e:  # surface server errors to the client
                context.set_code(grpc.StatusCode.INTERNAL)
                context.set_details(f"{type(e).__name__}: {e}")
                return pack({"error": str(e)})

        return grpc.unary_unary_rpc_method_handler(
            unary, request_deserializer=None, response_serializer=None)


def serve(port: int = 2222, task_index: int = 0, block: bool = True,
          ckpt_dir: str = "/tmp/tepdist_ckpt"):
    svc = TepdistService(task_index=task_index, ckpt_dir=ckpt_dir)
    server = grpc.server(
        futures.ThreadPoolExecutor(max_workers=8),
        options=[("grpc.max_receive_message_length", -1),
                 ("grpc.max_send_message_length", -1)])
    server.add_generic_rpc_handlers((_Handler(svc),))
    server.add_insecure_port(f"0.0.0.0:{port}")
    server.start()
    from tepdist_amd.config import get_env
    if get_env().debug:
        print(get_env().dump(), flush=True)   # PrintAllEnvs at startup
    print(f"[tepdist] server listening on :{port} (task {task_index})",
          flush=True)
    if block:
        server.wait_for_termination()
    return server, svc


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--port", type=int, default=2222)
    ap.add_argument("--task_index", type=int, default=0)
    ap.add_argument("--ckpt_dir", type=str, default="/tmp/tepdist_ckpt")
    args = ap.parse_args()
    serve(args.port, args.task_index, ckpt_dir=args.ckpt_dir)


if __name__ == "__main__":
    main()
