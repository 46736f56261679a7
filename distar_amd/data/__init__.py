from .fake_dataloader import FakeSLDataloader, FakeRLDataloader
from .adapter import Adapter
from .coordinator import Coordinator, Worker
