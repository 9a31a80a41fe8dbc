// devfilter — cgroup-v2 eBPF device-controller programs for egpu-hook.
//
// On cgroup-v2 hosts there is no devices.allow file: device access is
// enforced by BPF_PROG_TYPE_CGROUP_DEVICE programs attached to the
// container's cgroup (runc attaches one generated from the OCI config's
// linux.resources.devices list). Granting extra nodes therefore means
// REPLACING that program with one whose allowlist is the union of the
// container's rules and the GPU nodes — multi-attach cannot widen access
// because the kernel ANDs the verdicts of all attached programs.
//
// This header is shared by the hook binary and its self-test subcommands.
#pragma once

#include <linux/bpf.h>

#include <string>
#include <vector>

namespace devfilter {

// One device rule. maj/min < 0 means wildcard. access bits follow
// BPF_DEVCG_ACC_*: MKNOD=1, READ=2, WRITE=4.
struct DevRule {
  char type = 'a';  // 'c' char, 'b' block, 'a' all
  long long maj = -1;
  long long min = -1;
  unsigned access = 7;  // rwm
  bool allow = true;
};

// First-match-wins program over `rules`; unmatched access gets
// `default_allow`. Allow rules match when the requested access bits are a
// subset of the rule's; deny rules when they intersect (conservative).
std::vector<bpf_insn> build_prog(const std::vector<DevRule>& rules, bool default_allow);

// Load via BPF_PROG_LOAD (kernel verifier). Returns prog fd or -1 (err set).
int load_prog(const std::vector<bpf_insn>& insns, std::string* err);

// Attach `prog_fd` to the cgroup directory with BPF_F_ALLOW_MULTI, then
// detach every program that was attached there before the call (the
// replace step). Returns 0 or -1 (err set).
int replace_attached(const std::string& cgroup_dir, int prog_fd, std::string* err);

// Count of CGROUP_DEVICE programs directly attached to the cgroup (-1 on
// error). The hook only replaces when one exists: a cgroup with none is
// already unrestricted and attaching an allowlist would REMOVE access.
int query_attached_count(const std::string& cgroup_dir, std::string* err);

// Parse the `linux.resources.devices` array out of an OCI config.json body
// (machine-written JSON; minimal scanner consistent with egpu_hook.cpp).
// Returns rules in file order; `found` reports whether the array exists.
std::vector<DevRule> parse_oci_device_rules(const std::string& config, bool* found);

// unified-hierarchy ("0::<path>") cgroup path of a pid, joined to
// /sys/fs/cgroup. Empty when the pid has no v2 membership line.
std::string unified_cgroup_dir(long pid, const std::string& cgroup_root = "/sys/fs/cgroup");

// True when the host is pure cgroup-v2 (cgroup.controllers at the root and
// no v1 devices hierarchy) — the case where the eBPF grant is needed.
bool host_is_pure_v2(const std::string& cgroup_root = "/sys/fs/cgroup");

}  // namespace devfilter
