from glom_pytorch_amd.models.glom import Glom, GroupedFeedForward, ConsensusAttention

__all__ = ["Glom", "GroupedFeedForward", "ConsensusAttention"]
