"""pushcdn_amd — an MI355X-native publish/subscribe + direct-message CDN.

A from-scratch reimplementation of the capabilities of
EspressoSystems/Push-CDN (the reference, studied at /root/reference),
re-designed MI355X-first:

- one GPU-resident broker per MI355X device; subscription tables and the
  message pool live in HBM3E (288 GB/GPU)
- hot data-plane ops (batched BLS verify, topic-match, N-way fan-out,
  on-device capnp serde, direct-route lookup) are hand-written CDNA4 HIP
  kernels (csrc/hip/)
- broker<->broker routing rides RCCL collectives/P2P over xGMI
  (pushcdn_amd.parallel), one process per GPU via torch.distributed
- marshal, auth, discovery and the client library run on the host
  (control plane), wire-compatible with the reference's Cap'n Proto schema
"""

__version__ = "0.1.0"
