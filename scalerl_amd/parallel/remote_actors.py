"""Multi-node IMPALA: remote actor farms over the TCP control plane.

Completes the reference's dead ``hpc/`` design into a
working path: a learner node serves weights and receives rollout slots;
remote CPU-only nodes run actor processes against a local slot store and
ship filled slots back as raw tensor frames (parallel/cluster.py).

Topology:

  learner node                         remote node (CPU only)
  ───────────                          ─────────────────────
  RemoteSlotServer(WorkerServer)  ◄──  RemoteActorNode
    · serves config + flat weights      · pulls weights into the shared
    · "slot" frames → local store         CPU flat its actors alias
      + full_q (same consumption        · N actor procs (LocalPolicy) fill
      path as local actors)               a local RolloutStore
                                        · uploader ships slots, recycles

Bulk intra-node traffic stays on shared memory/RCCL; TCP carries only
cross-node slots (~29 MB per T=80,E=16 slot — one slot ≈ 1280 env steps,
so a 10 GbE link sustains ≈50k env-steps/s per node).
"""

from __future__ import annotations

import threading
from typing import Dict, List, Optional

import torch
import torch.multiprocessing as mp

from .cluster import RemoteWorkerCluster, WorkerServer
from .flat import FlatParams
from .rollout import RolloutStore, actor_loop

_SLOT_FIELDS = ("obs", "reward", "done", "last_action", "action", "logits",
                "episode_return")


class RemoteSlotServer:
    """Learner-side: accepts slot uploads into the learner's store.

    Remote slots are written into a RESERVED tail range of the store's
    slots (ids the local free/full queues never cycle), then announced on
    full_q — the learner's consumption path is identical for local and
    remote rollouts.
    """

    def __init__(self, store: RolloutStore, full_q, free_remote_q,
                 shared_flat: torch.Tensor, config: Dict, port: int = 0):
        self.store = store
        self.full_q = full_q
        self.free_remote_q = free_remote_q  # queue of reserved slot ids
        self.shared_flat = shared_flat
        self.server = WorkerServer(config, port=port,
                                   episode_callback=self._on_frame,
                                   retain_episodes=False)
        self.port = self.server.port
        self._lock = threading.Lock()

    def publish_weights(self) -> None:
        self.server.publish_weights(self.shared_flat)

    def _on_frame(self, header: Dict, tensors: List[torch.Tensor]) -> None:
        if header.get("type") != "slot":
            return
        slot = self.free_remote_q.get()  # blocks uploader until one frees
        with self._lock:
            fields = dict(zip(header["fields"], tensors))
            for name in header["fields"]:
                getattr(self.store, name)[slot].copy_(
                    fields[name].view_as(getattr(self.store, name)[slot]))
            self.full_q.put(slot)
        # NOTE: the learner recycles remote ids back onto free_remote_q
        # (ImpalaTrainer routes ids >= its local slot count there).

    def close(self) -> None:
        self.server.close()


def remote_actor_node(host: str, port: int, num_actors: int = 4,
                      stop_event=None, max_slots: int = 0,
                      weight_refresh_slots: int = 4):
    """Remote-node main: run actors locally, ship slots to the learner.

    The learner's config frame carries env/rollout geometry; the actor
    model weights land in a shared CPU flat that the forked actors alias
    (exactly the intra-node publication mechanism, over TCP instead of a
    learner memcpy).
    """
    from ..models.atari import AtariNet

    client = RemoteWorkerCluster(host, port)
    cfg = client.config
    E = cfg["envs_per_actor"]
    store = RolloutStore(2 * num_actors + 2, cfg["rollout_length"], E,
                         tuple(cfg["obs_shape"]), cfg["num_actions"],
                         lstm_layers=2,
                         lstm_hidden=cfg.get("lstm_hidden", 0))
    model = AtariNet(tuple(cfg["obs_shape"]), cfg["num_actions"],
                     use_lstm=cfg.get("lstm_hidden", 0) > 0)
    model.eval()
    flat = FlatParams(model, device="cpu", share=True)
    w = client.pull_weights()
    if w is not None:
        flat.load_from(w)

    ctx = mp.get_context("fork")
    free_q = ctx.SimpleQueue()
    full_q = ctx.SimpleQueue()
    stop = stop_event or ctx.Event()
    counter = ctx.Value("l", 0)
    env_spec = {"env_id": cfg["env_id"], "envs_per_actor": E,
                "seed": cfg.get("seed", 0) + 31 * client.worker_id}
    actors = []
    for i in range(num_actors):
        p = ctx.Process(target=actor_loop,
                        args=(i, env_spec, store, free_q, full_q, stop,
                              counter),
                        kwargs=dict(actor_model=model,
                                    seed=cfg.get("seed", 0)),
                        daemon=True)
        p.start()
        actors.append(p)
    for s in range(store.num_slots):
        free_q.put(s)

    shipped = 0
    try:
        while not stop.is_set() and (max_slots == 0 or shipped < max_slots):
            slot = full_q.get()
            tensors = [getattr(store, f)[slot].clone() for f in _SLOT_FIELDS]
            fields = list(_SLOT_FIELDS)
            if store.core_state is not None:
                tensors.append(store.core_state[slot].clone())
                fields.append("core_state")
            client.push_episode({"type": "slot", "fields": fields,
                                 "env_steps": store.rollout_length * E},
                                tensors)
            free_q.put(slot)
            shipped += 1
            if shipped % weight_refresh_slots == 0:
                w = client.pull_weights()
                if w is not None:
                    flat.load_from(w)
    finally:
        stop.set()
        for _ in actors:
            free_q.put(None)
        for p in actors:
            p.join(timeout=2.0)
            if p.is_alive():
                p.terminate()
        client.close()
    return shipped
