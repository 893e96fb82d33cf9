"""Generate tc-qdisc bandwidth-shaping playbooks
(parity: reference tools/create_bandwidth_playbook.py, generalized).

Used for QuantPipe-style constrained-bandwidth experiments on multi-node
deployments: emits Ansible playbooks that add/delete token-bucket rate
limits on each host group's NIC. (Intra-node xGMI links cannot be shaped
this way; for single-node experiments use the SEND_CONSTRAINT env with the
adaptive quantization policies instead - see runtime.py.)
"""
import argparse


def tc_command(bandwidth_mbit, action, dev):
    return (f"sudo tc qdisc {action} dev {dev} root tbf rate "
            f"{bandwidth_mbit}mbit burst 32kbit latency 20ms")


def write_playbook(path, host_group, command, task_name):
    with open(path, "w", encoding="utf-8") as script:
        script.write(f"- hosts: {host_group}\n  tasks:\n")
        script.write(f"    - name: {task_name}\n")
        script.write(f"      shell: {command}\n\n")
    print(f"wrote {path}")


def main():
    parser = argparse.ArgumentParser(
        description=__doc__,
        formatter_class=argparse.ArgumentDefaultsHelpFormatter)
    parser.add_argument("-b", "--bandwidths", type=str, default="5,10,50,100",
                        help="comma-separated rates in Mbit/s")
    parser.add_argument("-g", "--host-groups", type=str, default="all")
    parser.add_argument("--dev", type=str, default="eth0")
    args = parser.parse_args()
    for bw in (int(b) for b in args.bandwidths.split(",")):
        for group in args.host_groups.split(","):
            write_playbook(f"bw_{bw}mbps_20ms_add_{group}.yml", group,
                           tc_command(bw, "add", args.dev),
                           "add bandwidth limitation")
            write_playbook(f"bw_{bw}mbps_20ms_delete_{group}.yml", group,
                           tc_command(bw, "delete", args.dev),
                           "delete bandwidth limitation")


if __name__ == "__main__":
    main()
