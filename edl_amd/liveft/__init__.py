from .elastic import ElasticManager, ElasticStatus, LauncherInterface
