"""openembedding_amd — MI355X-native sparse-embedding training framework.

A from-scratch rebuild of the capabilities of 4paradigm/OpenEmbedding
(reference: /root/reference) as a GPU-resident sharded embedding engine:

- the reference's CPU parameter server (openembedding/variable, server, client)
  becomes a GPU open-addressed hash table / dense array table sharded across
  the GPUs of one node (CDNA4 HIP kernels, reference semantics of
  EmbeddingOptimizerVariable.h:242-297);
- the pico RPC fabric + master (reference pico-ps) becomes torch.distributed
  over RCCL/xGMI: sparse pull/push = all_to_all, dense grads = allreduce,
  control = broadcast/barrier (reference Communication.cpp -> torch.distributed);
- the Keras/TF glue (openembedding/tensorflow/exb.py) becomes a PyTorch module
  API in openembedding_amd.torch with the same surface: Embedding,
  distributed_optimizer, distributed_model, save/load_server_model, ...

Public layout:
  openembedding_amd.flags       -- process-level config (ref openembedding/__init__.py:8-41)
  openembedding_amd.torch       -- the user-facing module API (ref exb.py)
  openembedding_amd.core        -- variable engine (ref openembedding/variable/)
  openembedding_amd.parallel    -- sharded all-to-all engine (ref server+client collapsed)
  openembedding_amd.ops         -- HIP kernels (gfx950) + CPU reference ops
  openembedding_amd.models      -- DeepFM / WDL / xDeepFM / LR model zoo
  openembedding_amd.checkpoint  -- dump/load in the reference's shard-file layout
  openembedding_amd.serving     -- model registry + REST controller (ref entry/controller.cc)
  openembedding_amd.data        -- Criteo TSV pipeline (ref csv datasets + criteo_preprocess.py)
  openembedding_amd.inject      -- global nn.Embedding auto-patch (ref laboratory/inject)
"""

__version__ = "0.2.0"

version = __version__


class Flags:
    """Process-level configuration, mirroring the reference's
    ``openembedding.flags`` (reference openembedding/__init__.py:8-41).

    Attributes:
      config: YAML/JSON string with the EnvConfig-style tree (see config.py).
      master_endpoint: "host:port" of the rendezvous (torch.distributed init).
      bind_ip: local ip for rendezvous (unused intra-node; kept for parity).
      num_workers: world size; -1 = from env (torchrun).
      wait_num_servers: kept for API parity; the engine is always embedded
        (one shard per worker), matching the reference default -1
        (openembedding/__init__.py:24-27).
    """

    def __init__(self):
        self.config = ""
        self.master_endpoint = ""
        self.bind_ip = ""
        self.num_workers = -1
        self.wait_num_servers = -1


flags = Flags()


class Master:
    """Rendezvous master. The reference runs a TCP master daemon
    (entry/masterd.cc); intra-node torch.distributed only needs a TCPStore,
    which torch creates from MASTER_ADDR/MASTER_PORT. This class exists for
    API parity and standalone (world_size=1) use."""

    def __init__(self, bind_ip: str = "127.0.0.1", port: int = 0):
        self.endpoint = f"{bind_ip}:{port}" if port else bind_ip

    @property
    def running(self):
        return True


class Server:
    """Embedded parameter-server shard. In this framework every worker owns
    shard ``rank`` of every variable on its own GPU; there is no separate
    server process (the reference's embedded-server default,
    openembedding/__init__.py:57-76). Kept for API parity."""

    def __init__(self, master_endpoint: str = "", bind_ip: str = ""):
        self.master_endpoint = master_endpoint

    def join(self):
        return
