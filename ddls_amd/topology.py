"""Cluster topologies.

Reference: ``ddls/topologies/ramp.py:11`` (full-mesh RAMP optical cluster,
per-direction channels, per-transceiver bandwidth = total/num_comm_groups) and
``ddls/topologies/torus.py:10``.  Rebuilt without networkx: servers are dense
integer ids with "(c-r-s)" string names; the full mesh means every shortest
path is the direct hop [src, dst], which we exploit instead of caching
all-pairs shortest paths.
"""
from __future__ import annotations

from typing import Dict, List, Tuple, Union

from .devices import Channel, gen_channel_id


class _LazyChannelMap:
    """dict-like channel_id -> Channel that instantiates on first access."""

    def __init__(self, topo: "Ramp"):
        self._topo = topo
        self._channels: Dict[str, Channel] = {}

    def __getitem__(self, cid: str) -> Channel:
        ch = self._channels.get(cid)
        if ch is None:
            src, dst, num = cid.split("|")
            if (src not in self._topo.name_to_node
                    or dst not in self._topo.name_to_node
                    or not (0 <= int(num) < self._topo.num_channels)):
                raise KeyError(cid)
            ch = Channel(src, dst, int(num),
                         channel_bandwidth=self._topo.channel_bandwidth)
            self._channels[cid] = ch
        return ch

    def __contains__(self, cid: str) -> bool:
        try:
            self[cid]
            return True
        except KeyError:
            return False

    def values(self):
        return self._channels.values()  # instantiated channels only

    def keys(self):
        return self._channels.keys()

    def instantiated(self) -> Dict[str, Channel]:
        return self._channels


class _DirectCidCache:
    """(src_node, dst_node, channel_num) -> channel id string, grown lazily."""

    def __init__(self, topo: "Ramp"):
        self._topo = topo
        self._cache: Dict[Tuple[int, int, int], str] = {}

    def __getitem__(self, key: Tuple[int, int, int]) -> str:
        cid = self._cache.get(key)
        if cid is None:
            u, v, k = key
            cid = gen_channel_id(self._topo.node_names[u],
                                 self._topo.node_names[v], k)
            self._cache[key] = cid
        return cid


class Ramp:
    """RAMP: (communication_groups x racks x servers) full mesh."""

    def __init__(self,
                 num_communication_groups: int = 4,
                 num_racks_per_communication_group: int = 2,
                 num_servers_per_rack: int = 4,
                 num_channels: int = 1,
                 total_node_bandwidth: Union[int, float] = int(1.6e12),
                 intra_gpu_propagation_latency: float = 1.25e-6,
                 worker_io_latency: float = 100e-9):
        if num_racks_per_communication_group > num_communication_groups:
            raise ValueError(
                f"num_racks_per_communication_group ({num_racks_per_communication_group}) "
                f"must be <= num_communication_groups ({num_communication_groups})")
        self.num_communication_groups = num_communication_groups
        self.num_racks_per_communication_group = num_racks_per_communication_group
        self.num_servers_per_rack = num_servers_per_rack
        self.num_channels = num_channels
        self.total_node_bandwidth = total_node_bandwidth
        self.channel_bandwidth = total_node_bandwidth / num_communication_groups
        self.intra_gpu_propagation_latency = intra_gpu_propagation_latency
        self.worker_io_latency = worker_io_latency

        # server nodes: dense index <-> (c, r, s) coordinate <-> "c-r-s" name
        self.coords: List[Tuple[int, int, int]] = []
        for c in range(num_communication_groups):
            for r in range(num_racks_per_communication_group):
                for s in range(num_servers_per_rack):
                    self.coords.append((c, r, s))
        self.node_names = [f"{c}-{r}-{s}" for c, r, s in self.coords]
        self.name_to_node = {nm: i for i, nm in enumerate(self.node_names)}
        self.coord_to_node = {co: i for i, co in enumerate(self.coords)}
        self.num_nodes = len(self.coords)

        # per-direction channels on every (u, v) pair of the full mesh.
        # Instantiated LAZILY: a 1024-server mesh has ~1M directed channels
        # and RAMP jobs only ever touch the ones their flows use.
        self.channel_id_to_channel = _LazyChannelMap(self)

        # cached direct-link channel ids (full mesh -> every path is a single
        # hop); avoids string formatting on the per-flow hot path.  Grown on
        # demand (a dense precompute is O(nodes^2)).
        self.direct_cid = _DirectCidCache(self)

        # populated by the cluster environment
        self.node_workers: List[dict] = [dict() for _ in range(self.num_nodes)]
        self.worker_to_node: Dict[str, int] = {}
        self.worker_to_type: Dict[str, str] = {}
        self.worker_types: set = set()
        self.num_workers = 0

    @property
    def shape(self) -> Tuple[int, int, int]:
        return (self.num_communication_groups,
                self.num_racks_per_communication_group,
                self.num_servers_per_rack)

    def shortest_paths(self, src_node: int, dst_node: int) -> List[List[int]]:
        """Full mesh: one shortest path, the direct link."""
        return [[src_node, dst_node]]

    def channels_between(self, src_node: int, dst_node: int) -> List[str]:
        u, v = self.node_names[src_node], self.node_names[dst_node]
        return [gen_channel_id(u, v, k) for k in range(self.num_channels)]

    def reset_devices(self):
        for workers in self.node_workers:
            for w in workers.values():
                w.reset()
        for ch in self.channel_id_to_channel.values():
            ch.reset()


class Torus:
    """1D/2D/3D torus with per-direction channels
    (reference ``topologies/torus.py:10``); kept for the legacy generic cluster."""

    def __init__(self, x_dims: int, y_dims: int = 1, z_dims: int = 1,
                 num_channels: int = 1,
                 channel_bandwidth: Union[int, float] = int(1.25e9)):
        self.x_dims, self.y_dims, self.z_dims = x_dims, y_dims, z_dims
        self.num_channels = num_channels
        self.channel_bandwidth = channel_bandwidth
        self.coords = [(x, y, z) for x in range(x_dims) for y in range(y_dims)
                       for z in range(z_dims)]
        self.node_names = [f"{x}-{y}-{z}" for x, y, z in self.coords]
        self.name_to_node = {nm: i for i, nm in enumerate(self.node_names)}
        self.num_nodes = len(self.coords)

        def wrap(i, n):
            return i % n

        links = set()
        for (x, y, z) in self.coords:
            for dx, dy, dz in [(1, 0, 0), (0, 1, 0), (0, 0, 1)]:
                nx_, ny_, nz_ = wrap(x + dx, x_dims), wrap(y + dy, y_dims), wrap(z + dz, z_dims)
                if (nx_, ny_, nz_) == (x, y, z):
                    continue
                a = self.name_to_node[f"{x}-{y}-{z}"]
                b = self.name_to_node[f"{nx_}-{ny_}-{nz_}"]
                links.add((min(a, b), max(a, b)))
        self.channel_id_to_channel: Dict[str, Channel] = {}
        for a, b in links:
            for k in range(num_channels):
                for u, v in ((a, b), (b, a)):
                    ch = Channel(self.node_names[u], self.node_names[v], k,
                                 channel_bandwidth=channel_bandwidth)
                    self.channel_id_to_channel[ch.channel_id] = ch
        self.node_workers = [dict() for _ in range(self.num_nodes)]
        self.worker_to_node = {}
        self.worker_to_type = {}
        self.worker_types = set()
        self.num_workers = 0


def build_topology(topology_config: dict):
    ttype = topology_config["type"]
    if ttype == "ramp":
        return Ramp(**topology_config["kwargs"])
    if ttype == "torus":
        return Torus(**topology_config["kwargs"])
    raise ValueError(f"Unrecognised topology type {ttype}")
