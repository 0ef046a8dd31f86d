// croagent — native node agent CLI for the cro-amd operator.
//
// In cluster mode the controller runs off-node and reaches node hardware by
// exec'ing into the privileged node-agent pod (the role `nvidia-smi` + shell
// pipelines play in the reference, gpus.go:1040-1067).  This binary is what
// that exec invokes: fork-free KFD sysfs enumeration, per-device compute-pid
// attribution, PCI hot-plug, and the gfx950 health probe — each emitting
// JSON on stdout.
//
//   croagent list   [--sysroot /]            GPU inventory (KFD topology)
//   croagent cxl    [--sysroot /]            CXL.mem inventory (/sys/bus/cxl)
//   croagent pids   [--gpu-id N]             KFD compute processes
//   croagent probe  [--device N | --bdf B]   gfx950 MFMA/HBM health probe
//   croagent drain  --bdf 0000:5a:00.0       sysfs PCI remove
//   croagent rescan                          sysfs PCI rescan
//
// Build (cro_amd/hip/build.py):
//   hipcc --offload-arch=gfx950 -O2 croagent.cpp -L../hip -lcroprobe -o croagent

#include <dirent.h>
#include <unistd.h>

#include "../hip/croprobe.h"

#include <cinttypes>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <fstream>
#include <map>
#include <sstream>
#include <string>
#include <vector>

namespace {

std::string g_sysroot = "";

std::string path(const std::string& p) { return g_sysroot + p; }

bool read_file(const std::string& p, std::string* out) {
  std::ifstream f(path(p));
  if (!f) return false;
  std::stringstream ss;
  ss << f.rdbuf();
  *out = ss.str();
  return true;
}

bool write_file(const std::string& p, const std::string& data) {
  std::ofstream f(path(p));
  if (!f) return false;
  f << data;
  return bool(f);
}

std::vector<std::string> list_dir(const std::string& p) {
  std::vector<std::string> names;
  DIR* d = opendir(path(p).c_str());
  if (!d) return names;
  while (dirent* e = readdir(d)) {
    if (strcmp(e->d_name, ".") && strcmp(e->d_name, "..")) names.push_back(e->d_name);
  }
  closedir(d);
  return names;
}

std::map<std::string, unsigned long long> parse_properties(const std::string& text) {
  // values can exceed LLONG_MAX (unique_id is a full 64-bit fuse) — parse
  // per line with strtoull so one wide value cannot abort the whole file
  std::map<std::string, unsigned long long> props;
  std::istringstream in(text);
  std::string line;
  while (std::getline(in, line)) {
    std::istringstream ls(line);
    std::string key, value;
    if (ls >> key >> value) props[key] = strtoull(value.c_str(), nullptr, 10);
  }
  return props;
}

struct Gpu {
  int kfd_node = 0;
  long long gpu_id = 0;
  unsigned long long unique_id = 0;
  long long render_minor = 0;
  long long gfx_target = 0;
  long long vram_bytes = 0;
  std::string bdf;
  int card_index = -1;
  std::vector<long long> xgmi_peers;
};

std::string bdf_from_location(long long loc, long long domain) {
  char buf[32];
  snprintf(buf, sizeof(buf), "%04llx:%02llx:%02llx.%llx", domain, (loc >> 8) & 0xff,
           (loc >> 3) & 0x1f, loc & 0x7);
  return buf;
}

std::map<std::string, int> drm_cards_by_bdf() {
  std::map<std::string, int> out;
  for (const auto& entry : list_dir("/sys/class/drm")) {
    if (entry.rfind("card", 0) != 0 || entry.size() <= 4) continue;
    bool digits = true;
    for (size_t i = 4; i < entry.size(); ++i) digits &= bool(isdigit(entry[i]));
    if (!digits) continue;
    std::string uevent;
    if (!read_file("/sys/class/drm/" + entry + "/device/uevent", &uevent)) continue;
    std::istringstream in(uevent);
    std::string line;
    while (std::getline(in, line)) {
      const std::string kPrefix = "PCI_SLOT_NAME=";
      if (line.rfind(kPrefix, 0) == 0) out[line.substr(kPrefix.size())] = atoi(entry.c_str() + 4);
    }
  }
  return out;
}

std::vector<Gpu> enumerate_gpus() {
  std::vector<Gpu> gpus;
  auto cards = drm_cards_by_bdf();
  const std::string base = "/sys/class/kfd/kfd/topology/nodes";
  for (const auto& entry : list_dir(base)) {
    std::string props_text;
    if (!read_file(base + "/" + entry + "/properties", &props_text)) continue;
    auto props = parse_properties(props_text);
    if (props["simd_count"] == 0) continue;
    Gpu g;
    g.kfd_node = atoi(entry.c_str());
    g.unique_id = props["unique_id"];
    g.render_minor = (long long)props["drm_render_minor"];
    g.gfx_target = (long long)props["gfx_target_version"];
    g.bdf = bdf_from_location((long long)props["location_id"], (long long)props["domain"]);
    auto card = cards.find(g.bdf);
    if (card != cards.end()) g.card_index = card->second;
    std::string gpu_id_text;
    if (read_file(base + "/" + entry + "/gpu_id", &gpu_id_text)) g.gpu_id = atoll(gpu_id_text.c_str());
    for (const auto& bank : list_dir(base + "/" + entry + "/mem_banks")) {
      std::string btext;
      if (!read_file(base + "/" + entry + "/mem_banks/" + bank + "/properties", &btext)) continue;
      auto bprops = parse_properties(btext);
      if (bprops["heap_type"] == 1 || bprops["heap_type"] == 2) g.vram_bytes += (long long)bprops["size_in_bytes"];
    }
    for (const auto& link : list_dir(base + "/" + entry + "/io_links")) {
      std::string ltext;
      if (!read_file(base + "/" + entry + "/io_links/" + link + "/properties", &ltext)) continue;
      auto lprops = parse_properties(ltext);
      if (lprops["type"] == 11) g.xgmi_peers.push_back((long long)lprops["node_to"]);  // xGMI
    }
    gpus.push_back(g);
  }
  return gpus;
}

std::string device_id(const Gpu& g) {
  char buf[64];
  if (g.unique_id) {
    snprintf(buf, sizeof(buf), "GPU-%016llx", g.unique_id);
    return buf;
  }
  return "GPU-pci-" + g.bdf;
}

int cmd_list() {
  auto gpus = enumerate_gpus();
  printf("{\"gpus\":[");
  for (size_t i = 0; i < gpus.size(); ++i) {
    const Gpu& g = gpus[i];
    printf("%s{\"device_id\":\"%s\",\"kfd_node\":%d,\"gpu_id\":%lld,"
           "\"render_minor\":%lld,\"card_index\":%d,\"pci_bdf\":\"%s\","
           "\"vram_bytes\":%lld,\"gfx_target\":%lld,\"xgmi_peers\":[",
           i ? "," : "", device_id(g).c_str(), g.kfd_node, g.gpu_id,
           g.render_minor, g.card_index, g.bdf.c_str(), g.vram_bytes, g.gfx_target);
    for (size_t j = 0; j < g.xgmi_peers.size(); ++j)
      printf("%s%lld", j ? "," : "", g.xgmi_peers[j]);
    printf("]}");
  }
  printf("]}\n");
  return 0;
}

int cmd_cxl() {
  const std::string base = "/sys/bus/cxl/devices";
  printf("{\"memdevs\":[");
  bool first = true;
  for (const auto& entry : list_dir(base)) {
    if (entry.rfind("mem", 0) != 0 || entry.size() <= 3 || !isdigit(entry[3])) continue;
    std::string text;
    unsigned long long serial = 0, size = 0;
    long long numa = 0;
    if (read_file(base + "/" + entry + "/serial", &text))
      serial = strtoull(text.c_str(), nullptr, 16);
    if (read_file(base + "/" + entry + "/ram/size", &text))
      size = strtoull(text.c_str(), nullptr, 16);
    if (read_file(base + "/" + entry + "/numa_node", &text)) numa = atoll(text.c_str());
    std::string bdf;
    if (read_file(base + "/" + entry + "/device/uevent", &text)) {
      std::istringstream in(text);
      std::string line;
      while (std::getline(in, line)) {
        const std::string kPrefix = "PCI_SLOT_NAME=";
        if (line.rfind(kPrefix, 0) == 0) bdf = line.substr(kPrefix.size());
      }
    }
    char id[64];
    if (serial) snprintf(id, sizeof(id), "CXL-%016llx", serial);
    else snprintf(id, sizeof(id), "CXL-pci-%s", bdf.c_str());
    printf("%s{\"device_id\":\"%s\",\"memdev\":\"%s\",\"size_bytes\":%llu,"
           "\"numa_node\":%lld,\"pci_bdf\":\"%s\"}",
           first ? "" : ",", id, entry.c_str(), size, numa, bdf.c_str());
    first = false;
  }
  printf("]}\n");
  return 0;
}

int cmd_pids(long long gpu_id) {
  const std::string base = "/sys/class/kfd/kfd/proc";
  printf("{\"pids\":[");
  bool first = true;
  for (const auto& entry : list_dir(base)) {
    if (entry.empty() || !isdigit(entry[0])) continue;
    if (gpu_id >= 0) {
      std::string vram;
      char fname[64];
      snprintf(fname, sizeof(fname), "/vram_%lld", gpu_id);
      if (!read_file(base + "/" + entry + fname, &vram)) continue;
      if (atoll(vram.c_str()) <= 0) continue;
    }
    printf("%s%s", first ? "" : ",", entry.c_str());
    first = false;
  }
  printf("]}\n");
  return 0;
}

int cmd_drain(const std::string& bdf) {
  if (bdf.empty()) {
    fprintf(stderr, "drain requires --bdf\n");
    return 2;
  }
  if (!write_file("/sys/bus/pci/devices/" + bdf + "/remove", "1")) {
    fprintf(stderr, "failed to write pci remove for %s\n", bdf.c_str());
    return 1;
  }
  printf("{\"drained\":\"%s\"}\n", bdf.c_str());
  return 0;
}

int cmd_rescan() {
  if (!write_file("/sys/bus/pci/rescan", "1")) {
    fprintf(stderr, "failed to write pci rescan\n");
    return 1;
  }
  printf("{\"rescanned\":true}\n");
  return 0;
}

int resolve_device_by_bdf(const std::string& bdf) {
  int count = cro_probe_device_count();
  char buf[64];
  std::string want = bdf.substr(0, bdf.rfind('.'));  // match domain:bus:dev
  for (int dev = 0; dev < count; ++dev) {
    if (cro_probe_pci_bus_id(dev, buf, sizeof(buf)) == 0) {
      std::string got(buf);
      for (auto& c : got) c = tolower(c);
      if (got.rfind(want, 0) == 0) return dev;
    }
  }
  return -1;
}

int cmd_probe(int device) {
  CroProbeResult result;
  int rc = cro_probe_run(device, &result);
  printf("{\"ok\":%s,\"rc\":%d,\"mfma_f32_exact\":%s,\"hbm_gbps\":%.1f,"
         "\"bf16_tflops\":%.1f,\"vram_total\":%lld,\"vram_free\":%lld,"
         "\"gcn_arch\":\"%s\",\"msg\":\"%s\"}\n",
         result.ok ? "true" : "false", rc, result.mfma_f32_exact ? "true" : "false",
         result.hbm_gbps, result.bf16_tflops, result.vram_total, result.vram_free,
         result.gcn_arch, result.msg);
  return result.ok ? 0 : 1;
}

}  // namespace

int main(int argc, char** argv) {
  if (argc < 2) {
    fprintf(stderr, "usage: croagent <list|cxl|pids|probe|drain|rescan> [options]\n");
    return 2;
  }
  std::string cmd = argv[1];
  std::string bdf;
  long long gpu_id = -1;
  int device = 0;
  for (int i = 2; i < argc; ++i) {
    std::string arg = argv[i];
    if (arg == "--sysroot" && i + 1 < argc) g_sysroot = argv[++i];
    else if (arg == "--bdf" && i + 1 < argc) bdf = argv[++i];
    else if (arg == "--gpu-id" && i + 1 < argc) gpu_id = atoll(argv[++i]);
    else if (arg == "--device" && i + 1 < argc) device = atoi(argv[++i]);
  }
  if (cmd == "list") return cmd_list();
  if (cmd == "cxl") return cmd_cxl();
  if (cmd == "pids") return cmd_pids(gpu_id);
  if (cmd == "probe") {
    if (!bdf.empty()) {
      std::string lower = bdf;
      for (auto& c : lower) c = tolower(c);
      device = resolve_device_by_bdf(lower);
      if (device < 0) {
        fprintf(stderr, "no HIP device with bdf %s\n", bdf.c_str());
        return 1;
      }
    }
    return cmd_probe(device);
  }
  if (cmd == "drain") return cmd_drain(bdf);
  if (cmd == "rescan") return cmd_rescan();
  fprintf(stderr, "unknown command %s\n", cmd.c_str());
  return 2;
}
