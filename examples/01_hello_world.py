"""Hello world: deploy a function onto the cluster (or local driver) and
call it. Run with KT_LOCAL_MODE=true for the no-Kubernetes dev loop."""
import kubetorch_amd as kt


def hello(name: str = "world"):
    import socket

    return f"hello {name} from {socket.gethostname()}"


if __name__ == "__main__":
    remote = kt.fn(hello).to(kt.Compute(cpus=1))
    print(remote("MI355X"))
    # hot loop: edit `hello` above and re-run — warm pods reload in <1 s
    remote.teardown()
