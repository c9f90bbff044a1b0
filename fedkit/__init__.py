"""fedkit — an MI355X-native federated / consensus training engine.

Re-creation of the capabilities of SarodYatawatta/federated-pytorch-test
(reference mounted read-only at /root/reference), designed MI355X-first:

* each of the K federated clients is ONE process pinned to ONE MI355X GPU
  (`fedkit.parallel.comm.DistComm`), with its 1/K data shard resident in HBM;
* the per-block parameter-subset aggregation (FedAvg / FedProx / consensus
  ADMM with adaptive Barzilai-Borwein rho) is an RCCL all-reduce over xGMI
  (`fedkit.algos`), not an in-process tensor loop;
* the hot ops (conv3x3/1x1 implicit-GEMM on MFMA, BatchNorm+ELU fusions,
  fused losses, flat param pack/unpack) are hand-written CDNA4 HIP kernels
  in `csrc/`, exposed through `fedkit.ops`;
* `fedkit.optim.LBFGSNew` is the stochastic L-BFGS with Wolfe/Armijo line
  searches (reference: src/lbfgsnew.py) on flat fp32 state buffers.

Layer map (cf. /root/repo/SURVEY.md §1-§2):
  L1 optimizer   -> fedkit.optim.LBFGSNew
  L2 model zoo   -> fedkit.models  (Net, Net1, Net2, ResNet18/9, VAE, VAE-CL, CPC)
  L3 param ABI   -> fedkit.utils.paramvec
  L4 drivers     -> fedkit.parallel.runtime + entry scripts at the repo root
  comm           -> fedkit.parallel.comm   (LocalComm == reference semantics,
                                            DistComm == RCCL over xGMI)
  kernels        -> fedkit.ops + csrc/
"""

__version__ = "0.1.0"

from . import models, utils, optim, algos, parallel, data  # noqa: F401
