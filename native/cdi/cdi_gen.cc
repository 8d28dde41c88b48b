// k3samd-cdi-gen — Container Device Interface spec generator.
//
// The modern counterpart of the OCI-wrapper injection path: containerd
// >= 1.7 natively applies CDI specs, so nodes that prefer CDI over a
// RuntimeClass wrapper run
//
//   k3samd-cdi-gen --output /etc/cdi/amd.com-gpu.json
//
// and the device plugin (started with --use-cdi) returns
// `cdi_devices: ["amd.com/gpu=<id>"]` from Allocate instead of raw device
// specs. Mirrors `nvidia-ctk cdi generate` in the NVIDIA stack (the
// injection role the reference delegates to the container toolkit,
// /root/reference/README.md:57-69).
//
// Spec shape (CDI 0.6): one named device per physical GPU (by stable id
// and by ordinal) plus "all"; /dev/kfd rides in every device's edits since
// compute requires it.

#include <cstdio>
#include <cstring>
#include <fstream>
#include <sys/stat.h>
#include <sys/sysmacros.h>

#include "../common/json_writer.h"
#include "../common/version.h"
#include "../topology/kfd_topology.h"

namespace {

struct Node {
  std::string path;
  int major, minor;
};

bool stat_cdev(const std::string& p, int& major, int& minor) {
  struct stat st;
  if (::stat(p.c_str(), &st) != 0 || !S_ISCHR(st.st_mode)) return false;
  major = (int)major(st.st_rdev);
  minor = (int)minor(st.st_rdev);
  return true;
}

void emit_device_node(k3samd::JsonWriter& w, const Node& n) {
  w.begin_obj();
  w.key("path").value(n.path);
  w.key("type").value("c");
  w.key("major").value(n.major);
  w.key("minor").value(n.minor);
  w.key("permissions").value("rw");
  w.end_obj();
}

}  // namespace

int main(int argc, char** argv) {
  if (k3samd::handle_version_flag(argc, argv, "k3samd-cdi-gen")) return 0;
  std::string output;
  std::string kind = "amd.com/gpu";
  std::string dev_root = "/dev";
  for (int i = 1; i < argc; ++i) {
    if (!std::strcmp(argv[i], "--output") && i + 1 < argc) output = argv[++i];
    else if (!std::strcmp(argv[i], "--kind") && i + 1 < argc) kind = argv[++i];
    else if (!std::strcmp(argv[i], "--dev-root") && i + 1 < argc)
      dev_root = argv[++i];
    else {
      std::printf("k3samd-cdi-gen [--output F] [--kind amd.com/gpu] "
                  "[--dev-root /dev]\n");
      return !std::strcmp(argv[i], "--help") ? 0 : 2;
    }
  }

  auto topo = k3samd::enumerate_topology(k3samd::default_sysfs_root());

  Node kfd{"/dev/kfd", 10, 241};
  stat_cdev(dev_root + "/kfd", kfd.major, kfd.minor);

  auto gpu_nodes = [&](const k3samd::GpuDevice& g) {
    std::vector<Node> ns;
    Node rn{"/dev/dri/renderD" + std::to_string(g.drm_render_minor), 226,
            g.drm_render_minor};
    stat_cdev(dev_root + "/dri/renderD" + std::to_string(g.drm_render_minor),
              rn.major, rn.minor);
    ns.push_back(rn);
    if (g.card_index >= 0) {
      Node cn{"/dev/dri/card" + std::to_string(g.card_index), 226,
              g.card_index};
      stat_cdev(dev_root + "/dri/card" + std::to_string(g.card_index),
                cn.major, cn.minor);
      ns.push_back(cn);
    }
    return ns;
  };

  k3samd::JsonWriter w;
  w.begin_obj();
  w.key("cdiVersion").value("0.6.0");
  w.key("kind").value(kind);
  w.key("devices").begin_arr();
  auto emit_device = [&](const std::string& name,
                         const std::vector<const k3samd::GpuDevice*>& gpus) {
    w.begin_obj();
    w.key("name").value(name);
    w.key("containerEdits").begin_obj();
    w.key("deviceNodes").begin_arr();
    emit_device_node(w, kfd);
    for (auto* g : gpus)
      for (const Node& n : gpu_nodes(*g)) emit_device_node(w, n);
    w.end_arr();
    w.end_obj();
    w.end_obj();
  };
  std::vector<const k3samd::GpuDevice*> all;
  for (size_t i = 0; i < topo.gpus.size(); ++i) {
    const auto& g = topo.gpus[i];
    all.push_back(&g);
    emit_device(std::to_string(i), {&g});       // by ordinal
    emit_device(g.stable_id(), {&g});           // by stable id
  }
  if (!all.empty()) emit_device("all", all);
  w.end_arr();
  w.end_obj();

  std::string text = w.str() + "\n";
  if (output.empty()) {
    std::fputs(text.c_str(), stdout);
  } else {
    std::string tmp = output + ".tmp";
    std::ofstream f(tmp, std::ios::trunc);
    f << text;
    f.close();
    if (!f || std::rename(tmp.c_str(), output.c_str()) != 0) {
      std::fprintf(stderr, "cdi-gen: cannot write %s\n", output.c_str());
      return 1;
    }
  }
  return 0;
}
