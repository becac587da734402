from .base import SCIClient, FakeSCIClient
