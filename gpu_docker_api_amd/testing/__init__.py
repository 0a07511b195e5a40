"""Out-of-process protocol simulators for subsystems whose real servers
(dockerd, etcd) are not installable in this offline image: live-socket
integration targets for the docker driver and the etcd gateway store."""
