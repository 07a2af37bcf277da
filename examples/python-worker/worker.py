"""Echo worker example (mirrors examples/python-worker/worker.py in the
reference): attach an in-process worker to a running node, or use the Worker
runtime class in your own process."""
from cordum_amd.runtime.node import Node
from cordum_amd.runtime.worker import echo_handler

if __name__ == "__main__":
    node = Node().start()
    node.add_worker("python-echo-1", handler=echo_handler, topics=["job.default", "job.echo"])
    print("echo worker attached; node ticking (ctrl-c to exit)")
    import time
    while True:
        node.tick()
        time.sleep(0.1)
