from .store import Store, ClusterStore, InMemoryStore  # noqa: F401
from .tcp_store import TcpStore, TcpStoreServer, ClusterTcpStore  # noqa: F401
