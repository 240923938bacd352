from xotorch_amd.download.downloader import (  # noqa: F401
  ShardDownloader,
  NoopShardDownloader,
  new_shard_downloader,
)
