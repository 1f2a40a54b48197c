"""GroupBatchNorm2d — BatchNorm with statistics shared across a GPU group.

Counterpart of the reference ``apex.contrib.cudnn_gbn.GroupBatchNorm2d``
(apex/contrib/cudnn_gbn/batch_norm.py:85+). The reference goes through the
cuDNN-frontend GBN graph with peer_memory buffers for the cross-GPU stat
exchange; on MI355X the same semantics run on the wave64 Welford kernels
(csrc/syncbn.hip) with the per-channel stats merged over RCCL/xGMI —
functionally identical, channels_last memory format supported.
"""

import torch

from ...parallel.sync_batchnorm import SyncBatchNorm, create_syncbn_process_group


class GroupBatchNorm2d(SyncBatchNorm):
    def __init__(self, num_features, group_size=1, eps=1e-5, momentum=0.1, affine=True,
                 track_running_stats=True):
        process_group = create_syncbn_process_group(group_size) if group_size > 1 else None
        super().__init__(num_features, eps=eps, momentum=momentum, affine=affine,
                         track_running_stats=track_running_stats, process_group=process_group)
        self.group_size = group_size

    def forward(self, input):
        if input.dim() == 4 and input.is_contiguous(memory_format=torch.channels_last):
            x = input.permute(0, 2, 3, 1)
            saved = self.channel_last
            self.channel_last = True
            out = SyncBatchNorm.forward(self, x)
            self.channel_last = saved
            return out.permute(0, 3, 1, 2).contiguous(memory_format=torch.channels_last)
        return SyncBatchNorm.forward(self, input)
