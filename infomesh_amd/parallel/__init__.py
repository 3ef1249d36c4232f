"""Intra-node distributed fabric: one process per GPU over
torch.distributed (backend "nccl" = RCCL over xGMI on ROCm; "gloo" for
CPU tests). Replaces the reference's libp2p/Kademlia network layer
(infomesh/p2p/) for the 8-GPU single-node deployment — SURVEY.md §5.8.
"""
