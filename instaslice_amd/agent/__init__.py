from instaslice_amd.agent.daemonset import NodeAgent  # noqa: F401
