"""MegaScope training-time WebSocket server.

Reference: training/training_wsserver.py:71 + wiring training.py:1991-2023.
Rank (pp0, tp0, dp0) serves the frontend; each training step is gated on a
``run_training_step`` message carrying visualization / disturbance /
compressor configs, which are broadcast to every rank before the step so
all TP peers enter the activation gathers together.  Tapped activations
flow through a queue to a sender thread ({"type":"update", ...} messages,
wire format SURVEY.md §2.6).
"""

from __future__ import annotations

import json
import queue
import threading
from typing import Optional

import torch.distributed as dist

from ..core import parallel_state
from ..core.tensor_disturbance import get_disturbance
from ..core.tensor_tracer import enable_tensor_tracers
from ..utils.ws import WebSocketConnection, WebSocketServer
from .global_vars import get_args, get_tokenizer


def _apply_configs(configs: dict, report_func=None):
    tt = enable_tensor_tracers()
    args = get_args()
    tt.set_num_layers(args.num_layers)
    tt.tokenizer = get_tokenizer()
    tt.tt_flags.set_by_configs(configs.get("visualization_flags") or {})
    tt.set_compressor_configs(configs.get("compressor_config") or {})
    get_disturbance().set_by_configs(configs.get("disturbance_configs") or {})
    if report_func is not None:
        tt.set_report(report_func)


def follower_sync_configs():
    """Non-server ranks: receive the step configs broadcast."""
    holder = [None]
    dist.broadcast_object_list(holder, src=0)
    configs = holder[0] or {}
    _apply_configs(configs, report_func=None)


class TrainingWSServer:
    def __init__(self, port: int, args=None):
        self.port = port
        self.args = args or get_args()
        self._conn: Optional[WebSocketConnection] = None
        self._step_event = threading.Event()
        self._pending_configs: dict = {}
        self._data_queue: "queue.Queue" = queue.Queue()
        self._server = WebSocketServer(port=port)
        self._sender_thread = None

    # ------------------------------------------------------------------
    def start(self):
        self._server.start_in_thread(self._client_loop)
        self._sender_thread = threading.Thread(target=self._data_sender,
                                               daemon=True)
        self._sender_thread.start()
        print(f"[MegaScope] training WS server on port {self.port}; "
              "steps are gated on run_training_step messages", flush=True)

    def _client_loop(self, conn: WebSocketConnection):
        self._conn = conn
        conn.send({"type": "start",
                   "micro_batch_size": self.args.micro_batch_size,
                   "seq_length": self.args.seq_length,
                   "num_layers": self.args.num_layers})
        while conn.open:
            msg = conn.recv_message()
            if msg is None:
                break
            try:
                data = json.loads(msg)
            except json.JSONDecodeError:
                continue
            if data.get("type") == "run_training_step":
                self._pending_configs = data
                self._step_event.set()
            elif data.get("type") == "ping":
                conn.send({"type": "pong"})
        self._conn = None
        self._step_event.set()  # unblock a waiting trainer on disconnect

    def _data_sender(self):
        while True:
            item = self._data_queue.get()
            if item is None:
                return
            conn = self._conn
            if conn is not None and conn.open:
                try:
                    conn.send(item)
                except (ConnectionError, OSError):
                    pass

    # ------------------------------------------------------------------
    def wait_for_step_and_broadcast(self):
        """Called by the trainer each iteration: wait for the frontend's
        go-ahead (if one is connected), then broadcast configs."""
        if self._conn is not None:
            self._step_event.wait()
            self._step_event.clear()
        configs = {
            "visualization_flags": self._pending_configs.get(
                "visualization_flags"),
            "disturbance_configs": self._pending_configs.get(
                "disturbance_configs"),
            "compressor_config": self._pending_configs.get(
                "compressor_config"),
        }
        if dist.is_initialized() and dist.get_world_size() > 1:
            dist.broadcast_object_list([configs], src=0)
        _apply_configs(configs, report_func=self._data_queue.put)

    def step_finished(self, iteration: int, loss_dict: dict):
        conn = self._conn
        if conn is not None and conn.open:
            try:
                conn.send({"type": "finish", "iteration": iteration,
                           "loss": {k: float(v) for k, v in loss_dict.items()}})
            except (ConnectionError, OSError):
                pass

    def stop(self):
        self._server.stop()
        self._data_queue.put(None)
