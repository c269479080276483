#pragma once

#include <functional>
#include <map>
#include <memory>
#include <mutex>
#include <string>

namespace stf {

class ResourceBase {
 public:
  virtual ~ResourceBase() {}
};

class ResourceMgr {
 public:
  // Returns the resource under `name`, creating it with `make` on first use.
  template <typename T>
  T* LookupOrCreate(const std::string& name, std::function<T*()> make) {
    std::lock_guard<std::mutex> l(mu_);
    auto it = resources_.find(name);
    if (it == resources_.end()) {
      it = resources_.emplace(name, std::unique_ptr<ResourceBase>(make())).first;
    }
    return static_cast<T*>(it->second.get());
  }

 private:
  std::mutex mu_;
  std::map<std::string, std::unique_ptr<ResourceBase>> resources_;
};

void* NewResourceMgr();
void DeleteResourceMgr(void*);

}  // namespace stf
