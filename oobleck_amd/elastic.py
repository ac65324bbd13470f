# Elastic control plane — dependency-free restatement of the reference's
# master/agent wire protocol and lifecycle
# (/root/reference/oobleck/elastic/master.py, agent.py, message_util.py).
#
# The reference's own modules cannot run in this environment: they import
# deepspeed (logging), simple_parsing, asyncssh and aiofiles, none of which
# are installed and there is no network to install them — the same blocker
# class as the C++ planner's empty submodules.  This file restates the parts
# the kill-a-rank drill needs, keeping the WIRE PROTOCOL byte-identical
# (message_util.py:33-93: 1-byte request type; 4-byte little-endian length +
# pickle payload; 2-byte response = Response, RequestType):
#
#   * ElasticMaster (master.py): asyncio TCP server; REGISTER_AGENT ->
#     job args; PING -> PONG; FORWARD_RANK0_PORT -> broadcast the rank-0
#     worker's torch rendezvous port to every agent (master.py:226 + the
#     forward handler); agent DISCONNECTION -> broadcast
#     Response.RECONFIGURATION + the lost agent's identity to the
#     survivors (master.py:192-204 close_agent).
#   * ElasticAgent (agent.py): register, receive args, spawn one worker
#     process per "GPU" with an mp.Pipe (agent.py:140-176), forward the
#     rank-0 port upward (agent.py:180-188), and forward master events
#     (reconfiguration, port) down the pipes (agent.py:214-222 +
#     on_receive_worker_port).
#
# Deviations, stated: agents are identified by AGENT INDEX, not node IP
# (every agent of the single-host drill shares 127.0.0.1 — the reference
# keys its rank map by IP, which only works with one agent per host); the
# agent is synchronous (socket + threads) where the reference uses asyncio
# (the master here is asyncio like the reference); ssh launching (the
# master starting agents remotely) is out of scope — the drill starts
# agents directly, which is also how the reference's tests drive them.
from __future__ import annotations

import asyncio
import enum
import multiprocessing
import pickle
import socket
import threading
from dataclasses import dataclass, field


class RequestType(enum.Enum):
    UNDEFINED = 0
    LAUNCH_JOB = 1
    GET_DIST_INFO = 2
    REGISTER_AGENT = 3
    PING = 4
    FORWARD_RANK0_PORT = 5


class Response(enum.Enum):
    SUCCESS = 1
    FAILURE = 2
    PONG = 3
    RECONFIGURATION = 4
    FORWARD_RANK0_PORT = 5


@dataclass
class DistributionInfo:
    agent_ids: list  # agent identities (the reference: node IPs)
    world_size: int


@dataclass
class JobArgs:
    """The drill's stand-in for OobleckArguments: everything a worker
    needs to build its pipelines (the reference carries model/dataset
    names; here the synthetic model dims travel directly)."""
    num_agents: int
    workers_per_agent: int
    model_dims: dict = field(default_factory=dict)
    ranks_lists: list = field(default_factory=list)
    min_num_ranks: int = 1
    microbatches: int = 2
    result_dir: str = ""


# ---- wire helpers (sync side; byte-identical to message_util.py) ----------

def sock_send(sock: socket.socket, msg, need_pickle=True) -> None:
    if need_pickle:
        msg = pickle.dumps(msg)
    sock.sendall(len(msg).to_bytes(4, "little") + msg)


def sock_recv_exactly(sock: socket.socket, n: int) -> bytes:
    buf = b""
    while len(buf) < n:
        chunk = sock.recv(n - len(buf))
        if not chunk:
            raise EOFError("connection closed")
        buf += chunk
    return buf


def sock_recv(sock: socket.socket, need_pickle=True):
    n = int.from_bytes(sock_recv_exactly(sock, 4), "little")
    msg = sock_recv_exactly(sock, n)
    return pickle.loads(msg) if need_pickle else msg


def sock_send_request_type(sock: socket.socket, t: RequestType) -> None:
    sock.sendall(t.value.to_bytes(1, "little"))


def sock_recv_response(sock: socket.socket) -> tuple[Response, RequestType]:
    b = sock_recv_exactly(sock, 2)
    return Response(b[0]), RequestType(b[1])


# ---- master ----------------------------------------------------------------

class ElasticMaster:
    """master.py's daemon loop for one job: serve agents, fan out the
    rank-0 rendezvous port, broadcast reconfiguration on agent loss."""

    def __init__(self, args: JobArgs, port: int = 0):
        self.args = args
        self.port = port
        self._agents: dict[int, tuple[asyncio.StreamReader,
                                      asyncio.StreamWriter]] = {}
        self._server: asyncio.AbstractServer | None = None
        self._loop: asyncio.AbstractEventLoop | None = None
        self._ready = threading.Event()
        self._thread: threading.Thread | None = None

    # -- protocol handlers --
    async def _send_response(self, w, req: RequestType, resp: Response):
        w.write(bytes([resp.value, req.value]))
        await w.drain()

    async def _send(self, w, msg):
        data = pickle.dumps(msg)
        w.write(len(data).to_bytes(4, "little") + data)
        await w.drain()

    async def _recv(self, r):
        n = int.from_bytes(await r.readexactly(4), "little")
        return pickle.loads(await r.readexactly(n))

    async def _on_connected(self, r, w):
        try:
            t = RequestType(int.from_bytes(await r.readexactly(1), "little"))
        except (asyncio.IncompleteReadError, ConnectionResetError):
            return
        if t is RequestType.REGISTER_AGENT:
            agent_id: int = await self._recv(r)
            self._agents[agent_id] = (r, w)
            await self._send_response(w, RequestType.REGISTER_AGENT,
                                      Response.SUCCESS)
            await self._send(w, self.args)
            await self._agent_handler(agent_id)

    async def _agent_handler(self, agent_id: int):
        r, w = self._agents[agent_id]
        try:
            while True:
                t = RequestType(int.from_bytes(await r.readexactly(1),
                                               "little"))
                if t is RequestType.PING:
                    await self._send_response(w, RequestType.PING,
                                              Response.PONG)
                elif t is RequestType.FORWARD_RANK0_PORT:
                    port: int = await self._recv(r)
                    # broadcast to every agent (master.py's
                    # forward_rank0_port_handler)
                    for aid, (_, aw) in list(self._agents.items()):
                        await self._send_response(
                            aw, RequestType.UNDEFINED,
                            Response.FORWARD_RANK0_PORT)
                        await self._send(aw, port)
        except (asyncio.IncompleteReadError, ConnectionResetError, EOFError):
            await self._close_agent(agent_id)

    async def _close_agent(self, agent_id: int):
        # master.py:192-204: pop + broadcast reconfiguration with the lost
        # agent's identity
        if agent_id not in self._agents:
            return
        self._agents.pop(agent_id)
        for aid, (_, aw) in list(self._agents.items()):
            try:
                await self._send_response(aw, RequestType.UNDEFINED,
                                          Response.RECONFIGURATION)
                await self._send(aw, agent_id)
            except (ConnectionResetError, BrokenPipeError):
                pass

    # -- lifecycle --
    def start_in_thread(self) -> int:
        """Run the asyncio server on a daemon thread; returns the port."""
        def run():
            self._loop = asyncio.new_event_loop()
            asyncio.set_event_loop(self._loop)

            async def serve():
                self._server = await asyncio.start_server(
                    self._on_connected, "127.0.0.1", self.port)
                self.port = self._server.sockets[0].getsockname()[1]
                self._ready.set()
                async with self._server:
                    await self._server.serve_forever()
            try:
                self._loop.run_until_complete(serve())
            except asyncio.CancelledError:
                pass
        self._thread = threading.Thread(target=run, daemon=True)
        self._thread.start()
        self._ready.wait(10)
        return self.port


# ---- agent -----------------------------------------------------------------

def agent_main(agent_id: int, master_port: int, worker_fn) -> None:
    """agent.py's lifecycle, synchronous: register, spawn workers with
    pipes, forward the rank-0 port upward, forward master events down.
    worker_fn(local_rank, agent_id, child_pipe, args) is the worker entry
    (the reference's worker_main; injected so the drill supplies its own
    engine wiring)."""
    sock = socket.create_connection(("127.0.0.1", master_port), timeout=10)
    sock_send_request_type(sock, RequestType.REGISTER_AGENT)
    sock_send(sock, agent_id)
    resp, req = sock_recv_response(sock)
    assert resp is Response.SUCCESS and req is RequestType.REGISTER_AGENT
    args: JobArgs = sock_recv(sock)

    dist_info = DistributionInfo(
        agent_ids=list(range(args.num_agents)),
        world_size=args.num_agents * args.workers_per_agent)

    ctx = multiprocessing.get_context("spawn")
    workers = []
    for wi in range(args.workers_per_agent):
        local_rank = agent_id * args.workers_per_agent + wi
        pipe, child = ctx.Pipe()
        p = ctx.Process(target=worker_fn,
                        args=(local_rank, agent_id, child, args),
                        daemon=True)
        p.start()
        workers.append((pipe, p))
        pipe.send(("dist", dist_info))

    # worker -> master port forwarding (any worker may elect itself rank 0
    # after a reconfiguration, so watch every pipe).  One lock around the
    # two-part send: concurrent watcher threads must not interleave the
    # request byte with another message's length-prefixed payload.
    send_lock = threading.Lock()

    def pipe_watcher(pipe):
        try:
            while True:
                msg = pipe.recv()
                if msg[0] == "port_out":
                    with send_lock:
                        sock_send_request_type(
                            sock, RequestType.FORWARD_RANK0_PORT)
                        sock_send(sock, msg[1])
        except (EOFError, OSError):
            pass
    for pipe, _ in workers:
        threading.Thread(target=pipe_watcher, args=(pipe,),
                         daemon=True).start()

    # master -> workers event loop
    try:
        while True:
            resp, _req = sock_recv_response(sock)
            if resp is Response.FORWARD_RANK0_PORT:
                port = sock_recv(sock)
                for pipe, _ in workers:
                    try:
                        pipe.send(("port", port))
                    except (BrokenPipeError, OSError):
                        pass
            elif resp is Response.RECONFIGURATION:
                lost_agent = sock_recv(sock)
                for pipe, _ in workers:
                    try:
                        pipe.send(("lost", lost_agent))
                    except (BrokenPipeError, OSError):
                        pass
    except (EOFError, OSError):
        pass
    for _, p in workers:
        p.join(timeout=60)
