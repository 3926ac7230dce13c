// Native VersionedMap — the eventually-consistent replicated map behind
// broker state sync (the C++ port of the reference's
// cdn-broker/src/connections/versioned_map.rs; semantics mirrored by the
// Python implementation in pushcdn_amd/broker/versioned_map.py, which
// delegates here when the native core is loaded).
//
// Per-key u64 version + tombstones; diff() returns (and drains) local
// changes; merge() applies a remote delta last-writer-wins on version with
// ties broken by the conflict identity (larger wins).  Delta wire format is
// the same length-prefixed encoding the Python side documents:
//   [u32 n] then per record:
//   [u8 has_value][u64 version][u16 cid_len][cid][u32 key_len][key]
//   ([u32 val_len][val] iff has_value)

#pragma once
#include <cstdint>
#include <cstring>
#include <map>
#include <optional>
#include <string>
#include <vector>

namespace state {

struct Versioned {
    std::optional<std::string> value;  // nullopt = tombstone
    uint64_t version = 0;
    std::string conflict_id;
};

struct Changed {
    std::string key;
    std::optional<std::string> old_value;
    std::optional<std::string> new_value;
};

class VersionedMap {
  public:
    explicit VersionedMap(std::string local_cid) : local_cid_(std::move(local_cid)) {}

    void insert(const std::string& key, const std::string& value) {
        modify_local(key, value);
    }

    void remove(const std::string& key) {
        auto it = map_.find(key);
        if (it != map_.end() && it->second.value.has_value()) {
            modify_local(key, std::nullopt);
        }
    }

    std::optional<std::string> get(const std::string& key) const {
        auto it = map_.find(key);
        if (it == map_.end()) return std::nullopt;
        return it->second.value;
    }

    size_t size() const {
        size_t n = 0;
        for (const auto& [k, v] : map_)
            if (v.value.has_value()) ++n;
        return n;
    }

    // drain local changes; purge shipped tombstones
    std::map<std::string, Versioned> diff() {
        auto d = std::move(dirty_);
        dirty_.clear();
        for (const auto& [k, e] : d) {
            auto it = map_.find(k);
            if (it != map_.end() && !it->second.value.has_value()) map_.erase(it);
        }
        return d;
    }

    std::map<std::string, Versioned> get_full() const { return map_; }

    std::vector<Changed> merge(const std::map<std::string, Versioned>& remote) {
        std::vector<Changed> changed;
        for (const auto& [k, re] : remote) {
            auto it = map_.find(k);
            bool take = false;
            if (it == map_.end()) {
                take = true;
            } else if (re.version > it->second.version) {
                take = true;
            } else if (re.version == it->second.version &&
                       re.conflict_id > it->second.conflict_id) {
                take = true;
            }
            if (!take) continue;
            std::optional<std::string> old =
                it == map_.end() ? std::nullopt : it->second.value;
            if (!re.value.has_value()) {
                if (it != map_.end()) map_.erase(it);
            } else {
                map_[k] = re;
            }
            if (old != re.value) changed.push_back({k, old, re.value});
        }
        return changed;
    }

    // ---------------- delta (de)serialization ----------------

    static std::vector<uint8_t> serialize_delta(const std::map<std::string, Versioned>& d) {
        std::vector<uint8_t> out;
        auto put = [&](const void* p, size_t n) {
            const uint8_t* b = (const uint8_t*)p;
            out.insert(out.end(), b, b + n);
        };
        uint32_t n = (uint32_t)d.size();
        put(&n, 4);
        for (const auto& [k, e] : d) {
            uint8_t has = e.value.has_value() ? 1 : 0;
            put(&has, 1);
            put(&e.version, 8);
            uint16_t cl = (uint16_t)e.conflict_id.size();
            put(&cl, 2);
            put(e.conflict_id.data(), cl);
            uint32_t kl = (uint32_t)k.size();
            put(&kl, 4);
            put(k.data(), kl);
            if (has) {
                uint32_t vl = (uint32_t)e.value->size();
                put(&vl, 4);
                put(e.value->data(), vl);
            }
        }
        return out;
    }

    static bool deserialize_delta(const uint8_t* data, size_t len,
                                  std::map<std::string, Versioned>* out) {
        size_t off = 0;
        auto need = [&](size_t n) { return off + n <= len; };
        if (!need(4)) return false;
        uint32_t n;
        memcpy(&n, data, 4);
        off = 4;
        for (uint32_t i = 0; i < n; ++i) {
            if (!need(11)) return false;
            uint8_t has = data[off];
            uint64_t version;
            memcpy(&version, data + off + 1, 8);
            uint16_t cl;
            memcpy(&cl, data + off + 9, 2);
            off += 11;
            if (!need(cl)) return false;
            std::string cid((const char*)data + off, cl);
            off += cl;
            if (!need(4)) return false;
            uint32_t kl;
            memcpy(&kl, data + off, 4);
            off += 4;
            if (!need(kl)) return false;
            std::string key((const char*)data + off, kl);
            off += kl;
            Versioned e;
            e.version = version;
            e.conflict_id = std::move(cid);
            if (has) {
                if (!need(4)) return false;
                uint32_t vl;
                memcpy(&vl, data + off, 4);
                off += 4;
                if (!need(vl)) return false;
                e.value = std::string((const char*)data + off, vl);
                off += vl;
            }
            (*out)[std::move(key)] = std::move(e);
        }
        return true;
    }

  private:
    void modify_local(const std::string& key, std::optional<std::string> value) {
        auto it = map_.find(key);
        uint64_t version = (it == map_.end()) ? 1 : it->second.version + 1;
        Versioned e{std::move(value), version, local_cid_};
        map_[key] = e;
        dirty_[key] = std::move(e);
    }

    std::string local_cid_;
    std::map<std::string, Versioned> map_;
    std::map<std::string, Versioned> dirty_;
};

}  // namespace state
