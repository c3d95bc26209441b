"""nn_distributed_training_amd — an MI355X-native decentralized consensus
training framework.

Re-implements the capability surface of `javieryu/nn_distributed_training`
(DiNNO / DSGD / DSGT decentralized optimization over sparse communication
graphs) as a brand-new framework designed for AMD Instinct MI355X (gfx950):

* one process per GPU (``torch.distributed`` over RCCL/xGMI), with logical
  graph nodes packed onto ranks,
* per-rank node replicas stored as a single flat parameter *stack* ``[L, n]``
  so the per-node hot loops of the reference (Python loops over nodes and
  parameter tensors) become single HIP kernel launches batched over nodes,
* hand-written CDNA4 HIP kernels (MFMA + LDS tiling) for the compute hot path,
* neighbor parameter/dual exchange as batched RCCL point-to-point send/recv
  derived from the communication-graph adjacency (never a global all-reduce),
* all-gather only where the algorithm is global: consensus-error evaluation
  and dynamic-graph position exchange.

Layer map (mirrors the reference's 4-layer split):
  experiments/  YAML-driven drivers        (reference: experiments/*.py)
  optimizers/   DiNNO, DSGD, DSGT          (reference: optimizers/*.py)
  problems/     MNIST / density problems   (reference: problems/*.py)
  models/ ops/ parallel/ utils/ data/      primitives
"""

__version__ = "0.1.0"
