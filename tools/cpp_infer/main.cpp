// Standalone C++ inference app (the reference's PytorchToCpp equivalent,
// README.md:65-79 — in-repo this time).
//
// Loads a TorchScript model traced by export.py (jit_traced_model_cpu.pth /
// jit_traced_model_gpu.pth) plus an image, runs detection, prints the boxes
// and reports FPS over --iters runs.
//
//   ./helmet_infer -m jit_traced_model_gpu.pth -i image.jpg [-n 100]
//
// The app depends only on LibTorch: images are binary PPM (P6), the
// zero-dependency format every tool can emit (`convert img.jpg img.ppm`,
// or `python tools/cpp_infer/to_ppm.py img.jpg`). Preprocessing matches
// utils.imload: resize to 512x512 (bilinear), scale to [0,1], ImageNet
// normalize.
// When the traced GPU model embeds the native gfx950 ops (torch.ops.rthd.*,
// the default export.py GPU trace), pass the kernel extension with
// -k <path to real_time_helmet_detection_amd/ops/_C*.so>; it is dlopen'd
// before torch::jit::load so the rthd:: custom ops resolve.
#include <torch/script.h>
#include <ATen/hip/HIPContext.h>
#include <dlfcn.h>

#include <chrono>
#include <cstdio>
#include <cstring>
#include <iostream>
#include <string>
#include <vector>

static torch::Tensor read_ppm(const std::string& path) {
  FILE* f = fopen(path.c_str(), "rb");
  if (!f) throw std::runtime_error("cannot open image: " + path);
  char magic[3] = {};
  int w = 0, h = 0, maxv = 0;
  if (fscanf(f, "%2s", magic) != 1 || strcmp(magic, "P6") != 0)
    throw std::runtime_error("image must be binary PPM (P6): " + path);
  // skip comments/whitespace
  auto read_int = [&]() {
    int c;
    do {
      c = fgetc(f);
      if (c == '#') { while (c != '\n' && c != EOF) c = fgetc(f); }
    } while (isspace(c) || c == '#');
    int v = 0;
    while (isdigit(c)) { v = v * 10 + (c - '0'); c = fgetc(f); }
    return v;
  };
  w = read_int();
  h = read_int();
  maxv = read_int();
  if (w <= 0 || h <= 0 || maxv != 255)
    throw std::runtime_error("unsupported PPM: " + path);
  std::vector<unsigned char> buf((size_t)w * h * 3);
  if (fread(buf.data(), 1, buf.size(), f) != buf.size())
    throw std::runtime_error("truncated PPM: " + path);
  fclose(f);
  return torch::from_blob(buf.data(), {h, w, 3}, torch::kUInt8).clone();
}

static torch::Tensor load_image(const std::string& path, int imsize) {
  auto img = read_ppm(path);
  img = img.permute({2, 0, 1}).to(torch::kFloat32).div_(255.0).unsqueeze(0);
  img = at::upsample_bilinear2d(img, {imsize, imsize},
                                /*align_corners=*/false);
  const float mean[3] = {0.485f, 0.456f, 0.406f};
  const float stdv[3] = {0.229f, 0.224f, 0.225f};
  for (int c = 0; c < 3; ++c)
    img[0][c] = (img[0][c] - mean[c]) / stdv[c];
  return img;
}

static int run(int argc, char** argv) {
  std::string model_path, image_path, kernels_path;
  int iters = 100, imsize = 512;
  for (int i = 1; i < argc - 1; ++i) {
    if (!strcmp(argv[i], "-m")) model_path = argv[++i];
    else if (!strcmp(argv[i], "-i")) image_path = argv[++i];
    else if (!strcmp(argv[i], "-k")) kernels_path = argv[++i];
    else if (!strcmp(argv[i], "-n")) iters = atoi(argv[++i]);
    else if (!strcmp(argv[i], "-s")) imsize = atoi(argv[++i]);
  }
  if (model_path.empty() || image_path.empty()) {
    std::cerr << "usage: " << argv[0]
              << " -m model.pth -i image.ppm [-k rthd_ops.so] [-n iters]"
                 " [-s imsize]\n";
    return 1;
  }
  if (!kernels_path.empty()) {
    // the kernel extension links libtorch_python, which expects the CPython
    // symbols to be present in the process (normally exported by the python
    // binary) — provide them via libpython before loading the extension.
    dlopen("libpython3.10.so.1.0", RTLD_NOW | RTLD_GLOBAL);
    if (!dlopen(kernels_path.c_str(), RTLD_NOW | RTLD_GLOBAL))
      throw std::runtime_error(std::string("dlopen failed: ") + dlerror());
  }

  torch::jit::script::Module model = torch::jit::load(model_path);
  model.eval();
  const bool cuda = torch::cuda::is_available() &&
                    model_path.find("gpu") != std::string::npos;
  torch::Device device(cuda ? torch::kCUDA : torch::kCPU);
  model.to(device);

  auto img = load_image(image_path, imsize).to(device);

  torch::NoGradGuard ng;
  auto out = model.forward({img}).toTuple();
  auto boxes = out->elements()[0].toTensor().cpu();
  auto clss = out->elements()[1].toTensor().cpu();
  auto scores = out->elements()[2].toTensor().cpu();
  const char* names[] = {"hat", "person"};
  const int nshow = std::min<int64_t>(boxes.size(0), 20);
  printf("%lld detections (showing %d)\n", (long long)boxes.size(0), nshow);
  for (int i = 0; i < nshow; ++i) {
    const int cls = clss[i].item<int64_t>();
    printf("det %2d: %-7s score %.3f box [%7.1f %7.1f %7.1f %7.1f]\n", i,
           cls < 2 ? names[cls] : "?", scores[i].item<float>(),
           boxes[i][0].item<float>(), boxes[i][1].item<float>(),
           boxes[i][2].item<float>(), boxes[i][3].item<float>());
  }

  // FPS: warmup 10 then timed iters
  for (int i = 0; i < 10; ++i) model.forward({img});
  if (cuda) at::hip::device_synchronize();
  auto t0 = std::chrono::steady_clock::now();
  for (int i = 0; i < iters; ++i) model.forward({img});
  if (cuda) at::hip::device_synchronize();
  auto t1 = std::chrono::steady_clock::now();
  const double sec = std::chrono::duration<double>(t1 - t0).count();
  printf("%d iters in %.3f s -> %.1f FPS @ %dx%d (%s)\n", iters, sec,
         iters / sec, imsize, imsize, cuda ? "gpu" : "cpu");
  return 0;
}

int main(int argc, char** argv) {
  try {
    return run(argc, argv);
  } catch (const std::exception& e) {
    std::cerr << "helmet_infer: " << e.what() << "\n";
    return 1;
  }
}
