from .client import CoordClient
from .server import CoordServer
