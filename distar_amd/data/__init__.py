from .fake_dataloader import FakeSLDataloader, FakeRLDataloader
