from .core import (save, load, to_device, recur, collate, batchify,
                   process_dataset, make_optimizer, make_scheduler, resume,
                   Stats, model_tag_of)

__all__ = ['save', 'load', 'to_device', 'recur', 'collate', 'batchify',
           'process_dataset', 'make_optimizer', 'make_scheduler', 'resume',
           'Stats', 'model_tag_of']
