from .store import EventStore, EventStoreWriter  # noqa: F401
from .dataset import EventSRDataset, resolve_scale_pair  # noqa: F401
from .sequence import SequenceDataset  # noqa: F401
from .loader import (SequenceDataLoader, InferenceSequenceDataLoader,  # noqa: F401
                     make_event_loader, read_datalist, sequence_collate)
from .synthetic import (generate_events, write_synthetic_store,  # noqa: F401
                        make_synthetic_dataset)
