from .sharder import dirichlet_shards, iid_shards, make_shards  # noqa: F401
from .synthetic import ClientShard, make_client_dataset  # noqa: F401
