"""Client API: the reference xla::Client's TepDist surface
(client/client.h:68-77,164-170; connect via SERVER_IP/SERVER_PORT like
client_library.cc:142-148). TepdistSession is the convenience layer a
frontend uses: export a model's IR, build the plan once, stream training
steps (inputs as tensors, loss back), save/restore checkpoints."""

from __future__ import annotations

import os
from typing import Dict, Optional

import grpc
import torch

from tepdist_amd.rpc.server import SERVICE_NAME
from tepdist_amd.rpc.wire import pack, unpack


class TepdistClient:
    def __init__(self, address: Optional[str] = None):
        if address is None:
            ip = os.environ.get("SERVER_IP", "127.0.0.1")
            port = os.environ.get("SERVER_PORT", "2222")
            address = f"{ip}:{port}"
        self.channel = grpc.insecure_channel(
            address, options=[("grpc.max_receive_message_length", -1),
                              ("grpc.max_send_message_length", -1)])

    def _call(self, method: str, req: dict) -> dict:
        fn = self.channel.unary_unary(f"/{SERVICE_NAME}/{method}")
        resp = unpack(fn(pack(req)))
        if isinstance(resp, dict) and resp.get("error"):
            raise RuntimeError(f"server error in {method}: {resp['error']}")
        return resp

    # -- the reference method surface --------------------------------------

    def build_execution_plan(self, graph_json: str, num_devices: int = 1,
                             init_specs: dict = None) -> dict:
        return self._call("BuildExecutionPlan",
                          {"graph": graph_json, "num_devices": num_devices,
                           "init_specs": init_specs or {}})

    def execute_plan(self, handle: int,
                     inputs: Dict[str, torch.Tensor]) -> dict:
        return self._call("ExecutePlan", {"handle": handle, "inputs": inputs})

    def transfer_to_server_host(self, name: str, data: torch.Tensor,
                                variable: bool = False,
                                global_idx: int = -1) -> dict:
        return self._call("TransferToServerHost",
                          {"name": name, "data": data, "variable": variable,
                           "global_idx": global_idx})

    def fetch_resource_vars(self, names=None) -> Dict[str, torch.Tensor]:
        return self._call("FetchResourceVars", {"names": names})["vars"]

    def do_remote_save(self, max_to_keep: int = 5,
                       global_step: int = 0) -> dict:
        return self._call("DoRemoteSave", {"max_to_keep": max_to_keep,
                                           "global_step": global_step})

    def do_remote_restore(self, global_step: Optional[int] = None) -> dict:
        return self._call("DoRemoteRestore", {"global_step": global_step})

    def init_remote_comm(self, master_addr: str, master_port: int, rank: int,
                         world: int, join: bool = False) -> dict:
        """join=True makes the worker enter the process group now (blocks
        until every rank arrives — fan the call out in parallel)."""
        return self._call("InitRemoteComm",
                          {"master_addr": master_addr,
                           "master_port": master_port, "rank": rank,
                           "world": world, "join": join})

    def transfer_module_and_defctx(self, graph_json: str,
                                   def_tree: str = "") -> dict:
        return self._call("TransferModuleAndDefCtx",
                          {"graph": graph_json, "def_tree": def_tree})

    def dispatch_plan(self, plan: dict = None) -> dict:
        return self._call("DispatchPlan", {"plan": plan})

    def execute_remote_plan(self, handle: int, inputs=None) -> dict:
        return self._call("ExecuteRemotePlan",
                          {"handle": handle, "inputs": inputs or {}})


class TepdistSession:
    """Frontend convenience: capture -> plan -> step loop. Supports the
    reference client's async step pipelining (NUM_PARALLEL_RPC_STEPS,
    xla_ops.cc:617-634) and periodic lazy variable fetch
    (FETCH_RESOURCE_VAR_STEPS)."""

    def __init__(self, client: TepdistClient = None):
        import concurrent.futures as _cf
        import json as _json
        self.client = client or TepdistClient()
        self.handle: Optional[int] = None
        self.plan_info: Optional[dict] = None
        # VARIABLE_MAP_FILE_PATH (reference tf2xla/xla_compiler.cc:957):
        # preload a variable-name -> global-arg-index map to the server
        vmap = os.environ.get("VARIABLE_MAP_FILE_PATH")
        if vmap and os.path.exists(vmap):
            with open(vmap) as f:
                self.client._call("TransferVarArgMap",
                                  {"map": _json.load(f)})
        self._n_parallel = int(os.environ.get("NUM_PARALLEL_RPC_STEPS", "1"))
        self._fetch_every = int(os.environ.get("FETCH_RESOURCE_VAR_STEPS",
                                               "0"))
        self._pool = _cf.ThreadPoolExecutor(max_workers=max(
            self._n_parallel, 1))
        self._inflight = []
        self._step_no = 0
        self.last_vars: Optional[Dict[str, torch.Tensor]] = None

    def compile_graph(self, graph, num_devices: int = 1) -> dict:
        self.plan_info = self.client.build_execution_plan(
            graph.to_json(), num_devices)
        self.handle = self.plan_info["handle"]
        return self.plan_info

    def step(self, inputs: Dict[str, torch.Tensor]) -> float:
        """Synchronous when NUM_PARALLEL_RPC_STEPS<=1, else returns the
        loss of the oldest in-flight step while pipelining new ones."""
        self._step_no += 1
        if self._fetch_every and self._step_no % self._fetch_every == 0:
            self.last_vars = self.client.fetch_resource_vars()
        if self._n_parallel <= 1:
            r = self.client.execute_plan(self.handle, inputs)
            return float(list(r["outputs"].values())[0])
        self._inflight.append(self._pool.submit(
            self.client.execute_plan, self.handle, inputs))
        if len(self._inflight) < self._n_parallel:
            return float("nan")  # warm-up: no result yet
        r = self._inflight.pop(0).result()
        return float(list(r["outputs"].values())[0])

    def drain(self):
        out = []
        while self._inflight:
            r = self._inflight.pop(0).result()
            out.append(float(list(r["outputs"].values())[0]))
        return out
