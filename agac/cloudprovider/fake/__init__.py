from .backend import FakeAWSBackend, FakeELBv2, FakeGlobalAccelerator, FakeRoute53

__all__ = ["FakeAWSBackend", "FakeELBv2", "FakeGlobalAccelerator", "FakeRoute53"]
