from .base import Cloud, CloudError, new_cloud
from .kind import KindCloud
from .gcp import GCPCloud
from .aws import AWSCloud
